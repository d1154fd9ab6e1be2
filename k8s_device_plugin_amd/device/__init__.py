"""Device vendor registry.

The reference registers four vendors behind a common interface
(/root/reference/pkg/device/devices.go:20-101); this stack is MI355X-only by
design (BASELINE.json: "no multi-vendor dispatch"), so the registry holds the
single AMD module but keeps the same call points (MutateAdmission, CheckType,
GenerateResourceRequests, allocation success/failure patching) so the
scheduler and plugin code read like the reference architecture.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

from ..utils.kubeclient import KubeClient
from ..utils import nodelock
from ..utils.types import (
    BIND_PHASE_ANNO,
    BIND_PHASE_FAILED,
    BIND_PHASE_SUCCESS,
    BIND_TIME_ANNO,
    PodInfo,
)
from .amd import AMDDevices, AMD_DEVICE_TYPE, HANDSHAKE_ANNO, REGISTER_ANNO

log = logging.getLogger(__name__)

_devices: Dict[str, AMDDevices] = {}
# handshake anno -> register anno, per vendor (reference devices.go:28-32)
KNOWN_DEVICES: Dict[str, str] = {}


def init_devices(**config_overrides) -> None:
    """Initialise the vendor registry.  Keyword overrides (the reference's
    GlobalFlagSet, devices.go:93-101) set AMDConfig fields, e.g.
    ``init_devices(resource_name="amd.com/gpu", default_mem=1024)``; a repeat
    call with no overrides keeps the existing registry."""
    if _devices and not config_overrides:
        return
    from .amd import AMDConfig

    cfg = AMDConfig(**config_overrides) if config_overrides else AMDConfig()
    dev = AMDDevices(cfg)
    _devices[AMD_DEVICE_TYPE] = dev
    KNOWN_DEVICES[HANDSHAKE_ANNO] = REGISTER_ANNO


def get_devices() -> Dict[str, AMDDevices]:
    init_devices()
    return _devices


def pod_allocation_try_success(client: KubeClient, node_name: str, pod: PodInfo) -> None:
    """If every pending entry is consumed, mark success and unlock the node.

    Reference: pkg/device/devices.go:54-78.
    """
    from ..utils.types import IN_REQUEST_DEVICES, SUPPORT_DEVICES
    from ..utils.codec import decode_pod_devices

    refreshed = client.get_pod(pod.name, pod.namespace)
    remaining = decode_pod_devices(IN_REQUEST_DEVICES, refreshed.annotations)
    for sd in remaining.values():
        for ctr in sd:
            if ctr:
                return  # more containers still to allocate
    annos = {BIND_PHASE_ANNO: BIND_PHASE_SUCCESS}
    client.patch_pod_annotations(pod.name, pod.namespace, annos)
    nodelock.release_node_lock(client, node_name)


def pod_allocation_failed(client: KubeClient, node_name: str, pod: PodInfo) -> None:
    """Mark failed and unlock so the pod is rescheduled (devices.go:80-91)."""
    client.patch_pod_annotations(pod.name, pod.namespace, {BIND_PHASE_ANNO: BIND_PHASE_FAILED})
    nodelock.release_node_lock(client, node_name)
