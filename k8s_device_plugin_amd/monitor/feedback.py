"""Priority blocking / contention feedback (monitor -> interceptor).

Reference behavior (cmd/vGPUmonitor/feedback.go:197-255):
- decay each region's recent_kernel every tick;
- build a per-device table of ACTIVE containers by priority (a container is
  active while its recent_kernel is positive — its processes are launching);
- CheckBlocking: while any higher-priority container is active on a device,
  write recent_kernel = -1 into lower-priority containers sharing it (their
  launch hook spins until restored);
- CheckPriority: enforce the CU limit only when a device is contended
  (>1 active container) — utilization_switch 0 lets a lone container
  free-run past its core quota.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Set, Tuple

from .pathmon import ContainerEntry, PathMonitor

log = logging.getLogger(__name__)

ACTIVITY_THRESHOLD = 0  # recent_kernel above this => active


class FeedbackLoop:
    def __init__(self, pathmon: PathMonitor, soft_cores: bool = False):
        """soft_cores=True: enforce the CU limit only while the device is
        contended (the reference's default GPU_CORE_UTILIZATION_POLICY);
        False (our default): strict isolation, limit always enforced."""
        self.pathmon = pathmon
        self.soft_cores = soft_cores

    def observe_once(self) -> None:
        entries = self.pathmon.live_regions()
        # device uuid -> {priority -> [entry]}
        by_device: Dict[str, Dict[int, List[ContainerEntry]]] = {}
        activity: Dict[str, bool] = {}
        for e in entries:
            snap = e.region.snapshot()
            rk = e.region.get_recent_kernel()
            active = rk > ACTIVITY_THRESHOLD and bool(snap.procs)
            activity[e.key] = active
            # decay: an idle container's recent_kernel drifts to 0
            if rk > 0:
                e.region.set_recent_kernel(rk - 1)
            for uuid in snap.uuids:
                if not uuid:
                    continue
                by_device.setdefault(uuid, {}).setdefault(snap.priority, []).append(e)

        blocked: Set[str] = set()
        contended: Set[str] = set()
        for uuid, prio_map in by_device.items():
            active_prios = sorted(
                p for p, ents in prio_map.items()
                if any(activity[e.key] for e in ents))
            n_active = sum(
                1 for ents in prio_map.values() for e in ents if activity[e.key])
            if n_active > 1:
                for ents in prio_map.values():
                    for e in ents:
                        contended.add(e.key)
            if not active_prios:
                continue
            top = active_prios[0]  # 0 = high beats 1 = low
            for p, ents in prio_map.items():
                if p > top:
                    for e in ents:
                        blocked.add(e.key)

        for e in entries:
            if e.key in blocked:
                e.region.set_recent_kernel(-1)
            elif e.region.get_recent_kernel() < 0:
                e.region.set_recent_kernel(0)  # unblock
            if self.soft_cores:
                e.region.set_utilization_switch(1 if e.key in contended else 0)
            else:
                e.region.set_utilization_switch(1)
