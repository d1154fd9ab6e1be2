"""Container cache-dir scanner + GC.

Reference: cmd/vGPUmonitor/pathmonitor.go:82-106 — walk
``$HOOK_PATH/containers/<podUID>_<ctr>/``, mmap each ``*.cache`` region,
drop and delete directories whose pod has been gone > 300 s.
"""
from __future__ import annotations

import glob
import logging
import os
import shutil
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set

from .region import SharedRegion

log = logging.getLogger(__name__)

GC_GRACE_SECONDS = 300.0


@dataclass
class ContainerEntry:
    key: str            # "<podUID>_<ctr>"
    pod_uid: str
    container: str
    path: str
    region: Optional[SharedRegion] = None
    missing_since: float = 0.0


class PathMonitor:
    def __init__(self, hook_path: str, lib_path: Optional[str] = None):
        self.containers_dir = os.path.join(hook_path, "containers")
        self.entries: Dict[str, ContainerEntry] = {}
        self.lib_path = lib_path
        # scan() runs on the monitor loop while live_regions() is called
        # from gRPC worker threads (noderpc) and the metrics collector
        self._lock = threading.Lock()

    def scan(self, live_pod_uids: Set[str], now: Optional[float] = None) -> None:
        with self._lock:
            self._scan_locked(live_pod_uids, now)

    def _scan_locked(self, live_pod_uids: Set[str], now: Optional[float]) -> None:
        now = time.time() if now is None else now
        seen = set()
        for d in glob.glob(os.path.join(self.containers_dir, "*")):
            key = os.path.basename(d)
            if "_" not in key:
                continue
            seen.add(key)
            entry = self.entries.get(key)
            if entry is None:
                pod_uid, _, ctr = key.partition("_")
                entry = ContainerEntry(key=key, pod_uid=pod_uid, container=ctr,
                                       path=d)
                self.entries[key] = entry
            if entry.region is None:
                caches = glob.glob(os.path.join(d, "*.cache"))
                if caches:
                    try:
                        region = SharedRegion(caches[0], self.lib_path)
                        if region.valid:
                            entry.region = region
                        else:
                            region.close()
                    except (OSError, ValueError) as e:
                        log.debug("cannot map %s: %s", caches[0], e)
            # GC bookkeeping
            if entry.pod_uid in live_pod_uids:
                entry.missing_since = 0.0
            else:
                if entry.missing_since == 0.0:
                    entry.missing_since = now
                elif now - entry.missing_since > GC_GRACE_SECONDS:
                    log.info("GC container cache dir %s (pod gone)", d)
                    if entry.region is not None:
                        entry.region.close()
                    shutil.rmtree(d, ignore_errors=True)
                    del self.entries[key]
        # drop entries whose dir vanished
        for key in list(self.entries):
            if key not in seen:
                e = self.entries.pop(key)
                if e.region is not None:
                    e.region.close()

    def live_regions(self) -> List[ContainerEntry]:
        with self._lock:
            return [e for e in self.entries.values() if e.region is not None]
