"""Pending-pod matching and per-container request consumption.

The scheduler's Bind pre-writes the whole pod's device assignment into the
``devices-to-allocate`` annotation; the device plugin's Allocate then consumes
one container's slice per gRPC call by locating "the pod bound to my node in
phase allocating" and erasing the first non-empty container entry.

Reference behavior: /root/reference/pkg/util/util.go:51-76 (GetPendingPod),
:216-234 (GetNextDeviceRequest), :244-271 (EraseNextDeviceTypeFromAnnotation).
"""
from __future__ import annotations

from typing import Optional, Tuple

from .codec import decode_pod_devices, encode_pod_single_device
from .kubeclient import KubeClient
from .types import (
    ASSIGNED_NODE_ANNO,
    BIND_PHASE_ALLOCATING,
    BIND_PHASE_ANNO,
    BIND_TIME_ANNO,
    IN_REQUEST_DEVICES,
    ContainerDevices,
    ContainerSpec,
    PodInfo,
)


class PendingPodError(RuntimeError):
    pass


def get_pending_pod(client: KubeClient, node: str) -> PodInfo:
    for p in client.list_pods():
        if BIND_TIME_ANNO not in p.annotations:
            continue
        if p.annotations.get(BIND_PHASE_ANNO) != BIND_PHASE_ALLOCATING:
            continue
        if p.annotations.get(ASSIGNED_NODE_ANNO) == node:
            return p
    raise PendingPodError(f"no binding pod found on node {node}")


def get_next_device_request(
    dtype: str, pod: PodInfo
) -> Tuple[ContainerSpec, ContainerDevices]:
    """First container with a non-empty pending assignment for ``dtype``."""
    pdevices = decode_pod_devices(IN_REQUEST_DEVICES, pod.annotations)
    pd = pdevices.get(dtype)
    if pd is None:
        raise PendingPodError("device request not found")
    for ctridx, ctr_devices in enumerate(pd):
        if ctr_devices:
            if ctridx >= len(pod.containers):
                raise PendingPodError(
                    f"assignment index {ctridx} beyond pod containers"
                )
            return pod.containers[ctridx], ctr_devices
    raise PendingPodError("device request not found")


def erase_next_device_type_from_annotation(
    client: KubeClient, dtype: str, pod: PodInfo
) -> None:
    """Blank the first non-empty container entry and re-patch the annotation."""
    pdevices = decode_pod_devices(IN_REQUEST_DEVICES, pod.annotations)
    pd = pdevices.get(dtype)
    if pd is None:
        raise PendingPodError("erase device annotation not found")
    res = []
    found = False
    for ctr_devices in pd:
        if not found and ctr_devices:
            found = True
            res.append([])
        else:
            res.append(ctr_devices)
    new_val = encode_pod_single_device(res)
    client.patch_pod_annotations(
        pod.name, pod.namespace, {IN_REQUEST_DEVICES[dtype]: new_val}
    )
    pod.annotations[IN_REQUEST_DEVICES[dtype]] = new_val
