"""Resource manager: physical GPU inventory -> fake-device fan-out.

Reference: pkg/device-plugin/.../rm/devices.go:144-166 — each physical GPU
becomes ``device_split_count`` kubelet devices with IDs ``<UUID>-<i>``, and
memory/cores scaling is applied when advertising (register.go:96-162).
"""
from __future__ import annotations

import logging
from dataclasses import dataclass
from typing import Dict, List, Optional

from .. import MI355X_DEVICE_TYPE
from ..utils.types import DeviceInfo
from .kfd import PhysicalGPU

log = logging.getLogger(__name__)

MIB = 1024 * 1024


@dataclass
class FakeDevice:
    id: str           # "<uuid>-<i>"
    uuid: str         # physical uuid
    numa: int
    healthy: bool = True


class ResourceManager:
    def __init__(self, gpus: List[PhysicalGPU], split_count: int = 10,
                 memory_scaling: float = 1.0, cores_scaling: float = 1.0,
                 device_type: str = MI355X_DEVICE_TYPE,
                 replica_overrides: Optional[Dict[str, int]] = None):
        self.gpus = gpus
        self.split_count = max(1, split_count)
        self.memory_scaling = memory_scaling
        self.cores_scaling = cores_scaling
        self.device_type = device_type
        # per-device replica counts from the time-slicing config block
        # (reference rm/device_map.go:37-317); "*" matches every device
        self.replica_overrides = dict(replica_overrides or {})
        self.health: Dict[str, bool] = {g.uuid: True for g in gpus}

    def replicas_for(self, uuid: str) -> int:
        ov = self.replica_overrides
        n = ov.get(uuid) or ov.get("*") or self.split_count
        return max(1, int(n))

    def by_uuid(self, uuid: str) -> Optional[PhysicalGPU]:
        for g in self.gpus:
            if g.uuid == uuid:
                return g
        return None

    def fake_devices(self) -> List[FakeDevice]:
        out: List[FakeDevice] = []
        for g in self.gpus:
            for i in range(self.replicas_for(g.uuid)):
                out.append(FakeDevice(
                    id=f"{g.uuid}-{i}",
                    uuid=g.uuid,
                    numa=g.numa_node,
                    healthy=self.health.get(g.uuid, True),
                ))
        return out

    @staticmethod
    def uuid_of_fake(fake_id: str) -> str:
        """'GPU-abc-3' -> 'GPU-abc' (strip the trailing replica index)."""
        if "-" in fake_id:
            head, _, tail = fake_id.rpartition("-")
            if tail.isdigit():
                return head
        return fake_id

    def api_devices(self) -> List[DeviceInfo]:
        """Node-annotation inventory with scaling applied
        (reference getApiDevices, register.go:96-162)."""
        out = []
        for idx, g in enumerate(self.gpus):
            # partitioned cards advertise a distinct type so pods can pin or
            # avoid them via use/nouse-gputype (the reference's MIG
            # mixed-strategy analog, rm/device_map.go:95-118): in CPX mode
            # each XCD is its own KFD node with cu_count 32 and 1/8 the HBM
            part = getattr(g, "compute_partition", "SPX")
            dtype = self.device_type if part in ("", "SPX") \
                else f"{self.device_type}-{part}"
            out.append(DeviceInfo(
                id=g.uuid,
                count=self.replicas_for(g.uuid),
                devmem=int(g.mem_bytes / MIB * self.memory_scaling),
                devcore=int(100 * self.cores_scaling),
                type=dtype,
                numa=g.numa_node,
                health=self.health.get(g.uuid, True),
                index=idx,
            ))
        return out

    def set_health(self, uuid: str, healthy: bool) -> bool:
        old = self.health.get(uuid)
        self.health[uuid] = healthy
        return old != healthy
