"""Standalone fake metrics collector (reference
cmd/vGPUmonitor/testcollector/main.go): serves the monitor's metric families
with synthetic data on :8080 so dashboards/scrape configs can be developed
without a node.  Run: python -m k8s_device_plugin_amd.monitor.testcollector
"""
from __future__ import annotations

import argparse
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

from prometheus_client import CollectorRegistry, generate_latest
from prometheus_client.core import GaugeMetricFamily

GB = 1 << 30


class FakeCollector:
    """Two fake MI355X devices, two fake pods with quotas/usage."""

    def collect(self):
        host_mem = GaugeMetricFamily(
            "HostGPUMemoryUsage", "physical VRAM used (fake)",
            labels=["deviceidx", "deviceuuid"])
        host_util = GaugeMetricFamily(
            "HostCoreUtilization", "GPU busy percent (fake)",
            labels=["deviceidx", "deviceuuid"])
        use = GaugeMetricFamily(
            "vGPU_device_memory_usage_in_bytes", "per-container usage (fake)",
            labels=["podnamespace", "podname", "ctrname", "vdeviceid"])
        lim = GaugeMetricFamily(
            "vGPU_device_memory_limit_in_bytes", "per-container quota (fake)",
            labels=["podnamespace", "podname", "ctrname", "vdeviceid"])
        phase = time.time() % 60 / 60
        for i, uuid in enumerate(["GPU-fake-0", "GPU-fake-1"]):
            host_mem.add_metric([str(i), uuid], (100 + 80 * phase) * GB)
            host_util.add_metric([str(i), uuid], 35 + 60 * phase)
        for pod in ["bench-a", "bench-b"]:
            use.add_metric(["default", pod, "main", "0"], (20 + 50 * phase) * GB)
            lim.add_metric(["default", pod, "main", "0"], 72 * GB)
        return [host_mem, host_util, use, lim]


def main(argv=None):
    p = argparse.ArgumentParser("vgpu-testcollector")
    p.add_argument("--port", type=int, default=8080)
    args = p.parse_args(argv)
    registry = CollectorRegistry()
    registry.register(FakeCollector())

    class H(BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def do_GET(self):
            body = generate_latest(registry)
            self.send_response(200)
            self.send_header("Content-Type", "text/plain; version=0.0.4")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

    srv = ThreadingHTTPServer(("0.0.0.0", args.port), H)
    print(f"fake metrics on :{srv.server_address[1]}/metrics")
    srv.serve_forever()


if __name__ == "__main__":
    main()
