"""Monitor Prometheus metrics (the :9394 endpoint).

Reference: cmd/vGPUmonitor/metrics.go:65-258 — HostGPUMemoryUsage,
HostCoreUtilization (per physical device, from the host's view) +
vGPU_device_memory_usage_in_bytes / _limit_in_bytes and
Device_memory_desc_of_container (context/module/data/offset breakdown) per
container from the shared regions.  Host values come from sysfs
(mem_info_vram_used / gpu_busy_percent), not a vendor library.
"""
from __future__ import annotations

import logging
import os
from typing import List, Optional

from prometheus_client import CollectorRegistry, generate_latest, start_http_server
from prometheus_client.core import GaugeMetricFamily

from ..plugin.kfd import PhysicalGPU
from .pathmon import PathMonitor

log = logging.getLogger(__name__)


def _read_int(path: str) -> Optional[int]:
    try:
        with open(path) as f:
            return int(f.read().strip())
    except (OSError, ValueError):
        return None


class MonitorCollector:
    def __init__(self, pathmon: PathMonitor, gpus: List[PhysicalGPU],
                 drm_root: str = "/sys/class/drm", node_name: str = ""):
        self.pathmon = pathmon
        self.gpus = gpus
        self.drm_root = drm_root
        self.node_name = node_name or os.environ.get("NodeName", "")

    def collect(self):
        host_mem = GaugeMetricFamily(
            "HostGPUMemoryUsage", "host view: device memory used (bytes)",
            labels=["deviceidx", "deviceuuid"])
        host_core = GaugeMetricFamily(
            "HostCoreUtilization", "host view: GPU busy percent",
            labels=["deviceidx", "deviceuuid"])
        ctr_usage = GaugeMetricFamily(
            "vGPU_device_memory_usage_in_bytes", "container device usage",
            labels=["poduid", "ctrname", "vdeviceid", "deviceuuid"])
        ctr_limit = GaugeMetricFamily(
            "vGPU_device_memory_limit_in_bytes", "container device limit",
            labels=["poduid", "ctrname", "vdeviceid", "deviceuuid"])
        ctr_desc = GaugeMetricFamily(
            "Device_memory_desc_of_container", "container memory breakdown",
            labels=["poduid", "ctrname", "vdeviceid", "deviceuuid", "bucket"])
        ctr_scale = GaugeMetricFamily(
            "vGPU_arbitrated_core_scale",
            "node-arbitrated CU-throttle multiplier written to the region",
            labels=["poduid", "ctrname", "vdeviceid", "deviceuuid"])

        for g in self.gpus:
            dev_dir = os.path.join(self.drm_root, f"card{g.drm_card}", "device")
            used = _read_int(os.path.join(dev_dir, "mem_info_vram_used"))
            busy = _read_int(os.path.join(dev_dir, "gpu_busy_percent"))
            if used is not None:
                host_mem.add_metric([str(g.index), g.uuid], used)
            if busy is not None:
                host_core.add_metric([str(g.index), g.uuid], busy)

        for entry in self.pathmon.live_regions():
            try:
                snap = entry.region.snapshot()
            except (OSError, ValueError) as e:
                log.debug("snapshot failed for %s: %s", entry.key, e)
                continue
            for d in range(snap.num_devices):
                uuid = snap.uuids[d] if d < len(snap.uuids) else ""
                labels = [entry.pod_uid, entry.container, str(d), uuid]
                ctr_usage.add_metric(labels, snap.device_usage(d))
                ctr_limit.add_metric(labels, snap.limit[d])
                # context/module/data buckets aggregated over processes
                buckets = {"context": 0, "module": 0, "data": 0}
                L = entry.region.layout
                import struct as _struct
                for s in range(L["_max_procs"]):
                    base = L["procs"] + s * L["_proc_slot_size"]
                    pid = entry.region._i32(base + L["_proc_pid"])
                    if pid == 0:
                        continue
                    off = base + L["_proc_used"] + d * L["_devmem_size"]
                    ctx_v, mod_v, buf_v = _struct.unpack_from(
                        "<QQQ", entry.region._mm, off)
                    buckets["context"] += ctx_v
                    buckets["module"] += mod_v
                    buckets["data"] += buf_v
                for name, val in buckets.items():
                    ctr_desc.add_metric(labels + [name], val)
                try:
                    ctr_scale.add_metric(
                        labels, entry.region.get_monitor_scale(d))
                except (OSError, ValueError):
                    pass

        return [host_mem, host_core, ctr_usage, ctr_limit, ctr_desc, ctr_scale]


def metrics_text(collector: MonitorCollector) -> bytes:
    registry = CollectorRegistry()
    registry.register(collector)
    return generate_latest(registry)


def serve_metrics(collector: MonitorCollector, port: int = 9394):
    registry = CollectorRegistry()
    registry.register(collector)
    start_http_server(port, registry=registry)
