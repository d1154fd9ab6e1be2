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
import time
from typing import Callable, Dict, List, Optional, Set

from .arbiter import ScaleArbiter
from .pathmon import ContainerEntry, PathMonitor

log = logging.getLogger(__name__)

ACTIVITY_THRESHOLD = 0  # recent_kernel above this => active

class FeedbackLoop:
    def __init__(self, pathmon: PathMonitor, soft_cores: bool = False,
                 busy_reader: Optional[Callable[[str], int]] = None,
                 interval_s: float = 5.0):
        """soft_cores=True: enforce the CU limit only while the device is
        contended (the reference's default GPU_CORE_UTILIZATION_POLICY);
        False (our default): strict isolation, limit always enforced.
        busy_reader(uuid) -> device busy percent is kept for metrics and
        backward compatibility; the arbitration controller itself runs on
        the regions' token buckets (see _arbitrate) and needs no host
        utilization signal.  interval_s is this loop's period; it is
        published into every region so the limiter sizes its
        scale-freshness window to it."""
        self.pathmon = pathmon
        self.soft_cores = soft_cores
        self.busy_reader = busy_reader
        self.interval_s = interval_s
        # device uuid -> slow-start controller (see monitor/arbiter.py)
        self._arbiters: Dict[str, ScaleArbiter] = {}
        self._activity: Dict[str, bool] = {}

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
        self._activity = activity

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
            try:
                e.region.set_monitor_interval(self.interval_s)
            except Exception:
                pass  # v2 region without the field
            if e.key in blocked:
                e.region.set_recent_kernel(-1)
            elif e.region.get_recent_kernel() < 0:
                e.region.set_recent_kernel(0)  # unblock
            if self.soft_cores:
                e.region.set_utilization_switch(1 if e.key in contended else 0)
            else:
                e.region.set_utilization_switch(1)

        self._arbitrate(by_device)

    def _arbitrate(self, by_device) -> None:
        """One scale per device, written to every region holding it.

        Control law and dynamics: monitor/arbiter.py (token-bound
        median target with slow-start).  This method only gathers the
        (active, bound) observation per device and publishes the scale
        to every region holding it."""
        now = time.monotonic_ns()
        for uuid, prio_map in by_device.items():
            ents = [e for es in prio_map.values() for e in es]
            active = 0
            bound = 0
            targets = []
            for e in ents:
                try:
                    snap = e.region.snapshot()
                    dev = snap.uuids.index(uuid)
                except (ValueError, OSError) as exc:
                    log.debug("region read %s failed: %s", e.key, exc)
                    continue
                targets.append((e, dev))
                lim = snap.sm_limit[dev]
                if not (0 < lim < 100):
                    continue
                if not self._activity.get(e.key):
                    continue
                active += 1
                try:
                    if e.region.get_core_tokens(dev) <= 0:
                        bound += 1
                except (OSError, ValueError):
                    pass
            arb = self._arbiters.setdefault(uuid, ScaleArbiter())
            scale = arb.tick(active, bound)
            for e, dev in targets:
                try:
                    e.region.set_monitor_scale(dev, scale, now)
                except Exception as exc:
                    log.warning("scale write %s failed: %s", e.key, exc)
