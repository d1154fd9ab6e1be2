"""Scheduler Prometheus metrics.

Reference: /root/reference/cmd/scheduler/metrics.go:47-219 — a custom
collector walking InspectAllNodesUsage + the scheduled-pod cache, exporting
the same 10 metric families (GPUDeviceMemoryLimit/CoreLimit/MemoryAllocated/
SharedNum/CoreAllocated, nodeGPUOverview, nodeGPUMemoryPercentage,
vGPUPodsDeviceAllocated, vGPUMemoryPercentage, vGPUCorePercentage).
"""
from __future__ import annotations

from typing import TYPE_CHECKING

from prometheus_client import CollectorRegistry, generate_latest
from prometheus_client.core import GaugeMetricFamily

if TYPE_CHECKING:
    from .core import Scheduler


class SchedulerCollector:
    def __init__(self, scheduler: "Scheduler"):
        self.scheduler = scheduler

    def collect(self):
        s = self.scheduler
        dev_limit = GaugeMetricFamily(
            "GPUDeviceMemoryLimit", "device memory limit (bytes)",
            labels=["nodeid", "deviceuuid"])
        core_limit = GaugeMetricFamily(
            "GPUDeviceCoreLimit", "device CU percent capacity",
            labels=["nodeid", "deviceuuid"])
        mem_alloc = GaugeMetricFamily(
            "GPUDeviceMemoryAllocated", "device memory allocated (bytes)",
            labels=["nodeid", "deviceuuid"])
        shared_num = GaugeMetricFamily(
            "GPUDeviceSharedNum", "number of tasks sharing this device",
            labels=["nodeid", "deviceuuid"])
        core_alloc = GaugeMetricFamily(
            "GPUDeviceCoreAllocated", "device CU percent allocated",
            labels=["nodeid", "deviceuuid"])
        node_overview = GaugeMetricFamily(
            "nodeGPUOverview", "node GPU overview (memory MiB used)",
            labels=["nodeid", "deviceuuid", "devicetype"])
        node_mem_pct = GaugeMetricFamily(
            "nodeGPUMemoryPercentage", "node GPU memory allocation fraction",
            labels=["nodeid", "deviceuuid"])
        pods_alloc = GaugeMetricFamily(
            "vGPUPodsDeviceAllocated", "vGPU memory (bytes) allocated per pod-device",
            labels=["podnamespace", "nodename", "podname", "containeridx",
                    "deviceuuid", "deviceusedcore"])
        vgpu_mem_pct = GaugeMetricFamily(
            "vGPUMemoryPercentage", "pod-device fraction of device memory",
            labels=["podnamespace", "nodename", "podname", "containeridx", "deviceuuid"])
        vgpu_core_pct = GaugeMetricFamily(
            "vGPUCorePercentage", "pod-device CU percent",
            labels=["podnamespace", "nodename", "podname", "containeridx", "deviceuuid"])

        MIB = 1024 * 1024
        totals = {}
        for node_id, usage in s.inspect_all_nodes_usage().items():
            for d in usage.devices:
                totals[d.id] = d.totalmem
                dev_limit.add_metric([node_id, d.id], d.totalmem * MIB)
                core_limit.add_metric([node_id, d.id], d.totalcore)
                mem_alloc.add_metric([node_id, d.id], d.usedmem * MIB)
                shared_num.add_metric([node_id, d.id], d.used)
                core_alloc.add_metric([node_id, d.id], d.usedcores)
                node_overview.add_metric([node_id, d.id, d.type], d.usedmem)
                if d.totalmem:
                    node_mem_pct.add_metric([node_id, d.id], d.usedmem / d.totalmem)

        for p in s.pod_manager.list_pods():
            for single in p.devices.values():
                for ctridx, ctrdevs in enumerate(single):
                    for dev in ctrdevs:
                        labels = [p.namespace, p.node_id, p.name, str(ctridx), dev.uuid]
                        pods_alloc.add_metric(labels + [str(dev.usedcores)],
                                              dev.usedmem * MIB)
                        total = totals.get(dev.uuid, 0)
                        if total:
                            vgpu_mem_pct.add_metric(labels, dev.usedmem / total)
                        vgpu_core_pct.add_metric(labels, dev.usedcores)

        return [dev_limit, core_limit, mem_alloc, shared_num, core_alloc,
                node_overview, node_mem_pct, pods_alloc, vgpu_mem_pct, vgpu_core_pct]


def metrics_text(scheduler: "Scheduler") -> bytes:
    registry = CollectorRegistry()
    registry.register(SchedulerCollector(scheduler))
    return generate_latest(registry)
