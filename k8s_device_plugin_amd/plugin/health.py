"""GPU health checking.

Reference behavior: NVML Xid event watcher marking devices Unhealthy
(rm/health.go:42-189) and the DCU /dev/kfd open probe (dcu/server.go:225).
MI355X sources, richest first:
  1. /dev/kfd openable (node-level: KFD alive);
  2. per-GPU KFD sysfs presence (the topology node can vanish on a fallen
     GPU);
  3. amdgpu RAS error counters when exposed
     (/sys/class/drm/card<N>/device/ras/{ue_count,ce_count}) — uncorrected
     errors mark the device Unhealthy, the analog of critical Xids.
``DP_DISABLE_HEALTHCHECKS=all`` disables (reference health.go parity).
"""
from __future__ import annotations

import logging
import os
import threading
from typing import Callable, Optional

from .kfd import kfd_healthy
from .rm import ResourceManager

log = logging.getLogger(__name__)


def _read_int(path: str) -> Optional[int]:
    try:
        with open(path) as f:
            return int(f.read().strip())
    except (OSError, ValueError):
        return None


def device_healthy(gpu, drm_root: str = "/sys/class/drm",
                   kfd_dev: str = "/dev/kfd",
                   topology_root: str = "/sys/class/kfd/kfd/topology") -> bool:
    if not kfd_healthy(kfd_dev):
        return False
    node_dir = os.path.join(topology_root, "nodes", str(gpu.node_id))
    if os.path.isdir(os.path.join(topology_root, "nodes")) and not os.path.isdir(node_dir):
        return False
    ue = _read_int(os.path.join(drm_root, f"card{gpu.drm_card}", "device",
                                "ras", "ue_count"))
    if ue is not None and ue > 0:
        log.warning("GPU %s has %d uncorrected RAS errors", gpu.uuid, ue)
        return False
    return True


class HealthChecker:
    def __init__(self, rm: ResourceManager, on_change: Callable[[], None],
                 interval_s: float = 5.0, drm_root: str = "/sys/class/drm",
                 kfd_dev: str = "/dev/kfd",
                 topology_root: str = "/sys/class/kfd/kfd/topology"):
        self.rm = rm
        self.on_change = on_change
        self.interval_s = interval_s
        self.drm_root = drm_root
        self.kfd_dev = kfd_dev
        self.topology_root = topology_root
        self._stop = threading.Event()
        self._thread = None

    def check_once(self) -> bool:
        if os.environ.get("DP_DISABLE_HEALTHCHECKS", "") == "all":
            return False
        changed = False
        for gpu in self.rm.gpus:
            healthy = device_healthy(gpu, self.drm_root, self.kfd_dev,
                                     self.topology_root)
            if self.rm.set_health(gpu.uuid, healthy):
                log.warning("GPU %s health -> %s", gpu.uuid, healthy)
                changed = True
        return changed

    def run(self) -> None:
        while not self._stop.is_set():
            if self.check_once():
                self.on_change()
            self._stop.wait(self.interval_s)

    def start(self) -> None:
        self._thread = threading.Thread(target=self.run, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)
