"""Scheduler core: node registry sync, usage snapshots, Filter and Bind.

Reference behavior: /root/reference/pkg/scheduler/scheduler.go —
- ``register_from_node_annotations`` (scheduler.go:132-238): poll loop that
  decodes each node's register annotation into the node manager and drives
  the handshake state machine (Reported -> Requesting_<ts> -> Deleted_<ts>
  after 60 s of silence, which evicts the node's devices);
- ``get_nodes_usage`` (scheduler.go:247-310): registry snapshot minus the
  usage of every cached assigned pod;
- ``filter`` (scheduler.go:354-407): score all candidate nodes, pick the
  HIGHEST score (bin-packing), pre-patch the pod's assignment annotations;
- ``bind`` (scheduler.go:312-352): lock the node, set bind-phase=allocating
  + bind-time, POST the Binding; on failure set failed + unlock.
"""
from __future__ import annotations

import logging
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..device import KNOWN_DEVICES, get_devices, init_devices
from ..utils import nodelock
from ..utils.codec import decode_node_xgmi, decode_node_devices, encode_pod_devices, decode_pod_devices
from ..utils.kubeclient import KubeClient
from ..utils.types import (
    XGMI_ANNO,
    ASSIGNED_NODE_ANNO,
    ASSIGNED_TIME_ANNO,
    BIND_PHASE_ALLOCATING,
    BIND_PHASE_ANNO,
    BIND_PHASE_FAILED,
    BIND_TIME_ANNO,
    IN_REQUEST_DEVICES,
    SUPPORT_DEVICES,
    DeviceUsage,
    PodDevices,
    PodInfo,
)
from .score import NodeScore, NodeUsage, calc_score, pod_device_requests
from .state import NodeManager, PodManager, SchedNodeInfo, SchedPodInfo

log = logging.getLogger(__name__)

HANDSHAKE_TIME_FORMAT = "%Y.%m.%d %H:%M:%S"
NODE_HANDSHAKE_TIMEOUT_S = 60.0
REGISTER_POLL_INTERVAL_S = 15.0


@dataclass
class FilterResult:
    node_names: List[str] = field(default_factory=list)
    failed_nodes: Dict[str, str] = field(default_factory=dict)
    error: str = ""


@dataclass
class BindResult:
    error: str = ""


class Scheduler:
    def __init__(self, client: KubeClient):
        init_devices()
        self.client = client
        self.node_manager = NodeManager()
        self.pod_manager = PodManager()
        self.overview_status: Dict[str, NodeUsage] = {}
        self.cached_status: Dict[str, NodeUsage] = {}
        self._stop = threading.Event()
        self._filter_lock = threading.Lock()
        self._lock = threading.RLock()

    # ------------------------------------------------------------------
    # Node registry sync
    # ------------------------------------------------------------------
    def register_from_node_annotations_once(self, now: Optional[float] = None) -> None:
        now = time.time() if now is None else now
        try:
            nodes = self.client.list_nodes()
        except Exception as e:  # keep the poll loop alive
            log.error("nodes list failed: %s", e)
            return
        node_names = []
        for node in nodes:
            node_names.append(node.name)
            for handshake_anno, register_anno in KNOWN_DEVICES.items():
                reg = node.annotations.get(register_anno)
                if reg is None:
                    continue
                try:
                    nodedevices = decode_node_devices(reg)
                except Exception as e:
                    log.error("failed to decode node %s devices: %s", node.name, e)
                    continue
                if not nodedevices:
                    continue
                handshake = node.annotations.get(handshake_anno, "")
                if "Requesting" in handshake:
                    # the plugin has not answered our request yet
                    try:
                        former = time.mktime(
                            time.strptime(handshake.split("_", 1)[1], HANDSHAKE_TIME_FORMAT)
                        )
                    except (IndexError, ValueError):
                        former = 0.0
                    if now > former + NODE_HANDSHAKE_TIMEOUT_S:
                        known = self.node_manager.get_node(node.name)
                        if known is not None:
                            self.node_manager.rm_node_devices(
                                node.name, [d.id for d in known.devices]
                            )
                            log.info("node %s devices evicted (handshake timeout)", node.name)
                            self._patch_handshake(node.name, handshake_anno, "Deleted_", now)
                    continue
                if "Deleted" in handshake:
                    continue
                # Reported (fresh heartbeat): challenge again and ingest
                self._patch_handshake(node.name, handshake_anno, "Requesting_", now)
                info = SchedNodeInfo(id=node.name)
                for index, d in enumerate(nodedevices):
                    d.index = index
                    info.devices.append(d)
                xgmi_anno = node.annotations.get(XGMI_ANNO)
                if xgmi_anno:
                    try:
                        info.xgmi = decode_node_xgmi(xgmi_anno)
                    except Exception as e:
                        log.warning("bad xgmi anno on %s: %s", node.name, e)
                self.node_manager.add_node(node.name, info)
        self.get_nodes_usage(node_names)

    def _patch_handshake(self, node_name: str, anno: str, prefix: str, now: float) -> None:
        try:
            self.client.patch_node_annotations(
                node_name,
                {anno: prefix + time.strftime(HANDSHAKE_TIME_FORMAT, time.localtime(now))},
            )
        except Exception as e:
            log.error("patch node %s handshake failed: %s", node_name, e)

    def register_loop(self) -> None:
        while not self._stop.is_set():
            self.register_from_node_annotations_once()
            self.rebuild_pod_cache()
            self._stop.wait(REGISTER_POLL_INTERVAL_S)

    def stop(self) -> None:
        self._stop.set()

    # ------------------------------------------------------------------
    # Pod cache rebuild (the informer-callback analogue; also crash recovery)
    # ------------------------------------------------------------------
    def ingest_assigned_pod(self, pod: PodInfo) -> None:
        """(Re)ingest any pod carrying an assignment (scheduler.go:73-126)."""
        node_id = pod.annotations.get(ASSIGNED_NODE_ANNO)
        if not node_id:
            return
        devices = decode_pod_devices(SUPPORT_DEVICES, pod.annotations)
        self.pod_manager.add_pod(pod, node_id, devices)

    def rebuild_pod_cache(self) -> None:
        """Reconcile the pod cache against the API: (re)ingest assigned
        pods AND evict vanished/completed ones (the reference's informer
        onAddPod/onDelPod pair, scheduler.go:73-126 — without eviction a
        deleted pod's usage would leak until restart)."""
        try:
            pods = self.client.list_pods()
        except Exception as e:
            log.error("pod list failed: %s", e)
            return
        live = set()
        for pod in pods:
            if pod.phase in ("Succeeded", "Failed"):
                continue
            live.add(pod.uid)
            self.ingest_assigned_pod(pod)
        for cached in self.pod_manager.list_pods():
            if cached.uid not in live:
                self.pod_manager.del_pod_by_uid(cached.uid)

    # ------------------------------------------------------------------
    # Usage snapshot
    # ------------------------------------------------------------------
    def get_nodes_usage(
        self, node_names: Optional[List[str]] = None
    ) -> Tuple[Dict[str, NodeUsage], Dict[str, str]]:
        with self._lock:
            overall: Dict[str, NodeUsage] = {}
            failed: Dict[str, str] = {}
            for node_id, node in self.node_manager.list_nodes().items():
                usage = NodeUsage(xgmi=node.xgmi)
                for d in node.devices:
                    usage.devices.append(
                        DeviceUsage(
                            id=d.id,
                            index=d.index,
                            used=0,
                            count=d.count,
                            usedmem=0,
                            totalmem=d.devmem,
                            totalcore=d.devcore,
                            usedcores=0,
                            type=d.type,
                            numa=d.numa,
                            health=d.health,
                        )
                    )
                overall[node_id] = usage
            for p in self.pod_manager.list_pods():
                node = overall.get(p.node_id)
                if node is None:
                    continue
                for single in p.devices.values():
                    for ctrdevs in single:
                        for udev in ctrdevs:
                            for d in node.devices:
                                if d.id == udev.uuid:
                                    d.used += 1
                                    d.usedmem += udev.usedmem
                                    d.usedcores += udev.usedcores
            self.overview_status = overall
            if node_names is None:
                self.cached_status = dict(overall)
                return overall, failed
            cached: Dict[str, NodeUsage] = {}
            for name in node_names:
                if name in overall:
                    cached[name] = overall[name]
                else:
                    failed[name] = "node unregistered"
            self.cached_status = cached
            return cached, failed

    def inspect_all_nodes_usage(self) -> Dict[str, NodeUsage]:
        return self.overview_status

    # ------------------------------------------------------------------
    # Filter / Bind
    # ------------------------------------------------------------------
    def filter(self, pod: PodInfo, node_names: List[str]) -> FilterResult:
        # kube-scheduler runs one scheduling cycle at a time, but the HTTP
        # server is threaded: serialize snapshot->score->commit so a custom
        # client can never over-commit a device
        with self._filter_lock:
            return self._filter_locked(pod, node_names)

    def _filter_locked(self, pod: PodInfo, node_names: List[str]) -> FilterResult:
        nums = pod_device_requests(pod)
        total = sum(int(k.nums) for n in nums for k in n.values())
        if total == 0:
            return FilterResult(node_names=list(node_names))
        self.pod_manager.del_pod(pod)
        usage, failed = self.get_nodes_usage(list(node_names))
        scores = calc_score(usage, nums, pod.annotations)
        if not scores:
            failed = dict(failed)
            for n in node_names:
                failed.setdefault(n, "no fitting device")
            return FilterResult(failed_nodes=failed)
        scores.sort(key=lambda s: s.score)
        best = scores[-1]  # highest score = busiest fitting node (binpack)
        annotations = {
            ASSIGNED_NODE_ANNO: best.node_id,
            ASSIGNED_TIME_ANNO: str(int(time.time())),
        }
        annotations.update(encode_pod_devices(IN_REQUEST_DEVICES, best.devices))
        annotations.update(encode_pod_devices(SUPPORT_DEVICES, best.devices))
        self.pod_manager.add_pod(pod, best.node_id, best.devices)
        try:
            self.client.patch_pod_annotations(pod.name, pod.namespace, annotations)
            pod.annotations.update(annotations)
        except Exception as e:
            self.pod_manager.del_pod(pod)
            return FilterResult(error=str(e))
        return FilterResult(node_names=[best.node_id])

    def bind(self, pod_name: str, pod_namespace: str, node_name: str) -> BindResult:
        try:
            nodelock.lock_node(self.client, node_name)
        except Exception as e:
            return BindResult(error=f"node lock failed: {e}")
        try:
            self.client.patch_pod_annotations(
                pod_name,
                pod_namespace,
                {
                    BIND_PHASE_ANNO: BIND_PHASE_ALLOCATING,
                    BIND_TIME_ANNO: str(int(time.time())),
                },
            )
            self.client.bind_pod(pod_name, pod_namespace, node_name)
            return BindResult()
        except Exception as e:
            log.error("bind %s/%s to %s failed: %s", pod_namespace, pod_name, node_name, e)
            try:
                self.client.patch_pod_annotations(
                    pod_name, pod_namespace, {BIND_PHASE_ANNO: BIND_PHASE_FAILED}
                )
            finally:
                try:
                    nodelock.release_node_lock(self.client, node_name)
                except Exception:
                    pass
            return BindResult(error=str(e))
