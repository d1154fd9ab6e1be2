"""In-memory scheduler state: node and pod managers.

Reference: /root/reference/pkg/scheduler/nodes.go:50-116 and pods.go:37-72 —
mutex-guarded maps rebuilt from annotations at any time ("annotations are the
database").
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..utils.types import DeviceInfo, PodDevices, PodInfo


@dataclass
class SchedNodeInfo:
    id: str
    devices: List[DeviceInfo] = field(default_factory=list)
    # uuid -> xGMI peer uuids (from amd.io/node-xgmi; may be empty)
    xgmi: Dict[str, List[str]] = field(default_factory=dict)


@dataclass
class SchedPodInfo:
    namespace: str
    name: str
    uid: str
    node_id: str
    devices: PodDevices = field(default_factory=dict)


class NodeManager:
    def __init__(self):
        self._lock = threading.RLock()
        self.nodes: Dict[str, SchedNodeInfo] = {}

    def add_node(self, node_id: str, info: SchedNodeInfo) -> None:
        """Merge: devices already known are refreshed in place; new ones are
        appended (reference scheduler.go:196-232 + nodes.go:60-86)."""
        with self._lock:
            cur = self.nodes.get(node_id)
            if cur is None:
                self.nodes[node_id] = SchedNodeInfo(
                    id=node_id, devices=list(info.devices),
                    xgmi=dict(info.xgmi))
                return
            if info.xgmi:
                cur.xgmi = dict(info.xgmi)
            for d in info.devices:
                for existing in cur.devices:
                    if existing.id == d.id:
                        existing.devmem = d.devmem
                        existing.devcore = d.devcore
                        existing.health = d.health
                        break
                else:
                    cur.devices.append(d)

    def rm_node_devices(self, node_id: str, device_ids: List[str]) -> None:
        with self._lock:
            cur = self.nodes.get(node_id)
            if cur is None:
                return
            cur.devices = [d for d in cur.devices if d.id not in set(device_ids)]
            if not cur.devices:
                del self.nodes[node_id]

    def get_node(self, node_id: str) -> Optional[SchedNodeInfo]:
        with self._lock:
            return self.nodes.get(node_id)

    def list_nodes(self) -> Dict[str, SchedNodeInfo]:
        with self._lock:
            return dict(self.nodes)


class PodManager:
    def __init__(self):
        self._lock = threading.RLock()
        self.pods: Dict[str, SchedPodInfo] = {}  # keyed by uid

    def add_pod(self, pod: PodInfo, node_id: str, devices: PodDevices) -> None:
        with self._lock:
            self.pods[pod.uid] = SchedPodInfo(
                namespace=pod.namespace,
                name=pod.name,
                uid=pod.uid,
                node_id=node_id,
                devices=devices,
            )

    def del_pod(self, pod: PodInfo) -> None:
        with self._lock:
            self.pods.pop(pod.uid, None)

    def del_pod_by_uid(self, uid: str) -> None:
        with self._lock:
            self.pods.pop(uid, None)

    def list_pods(self) -> List[SchedPodInfo]:
        with self._lock:
            return list(self.pods.values())
