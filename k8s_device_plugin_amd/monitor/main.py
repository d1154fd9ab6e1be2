"""vGPUmonitor daemon: metrics + feedback + host-pid mapping.

Reference: cmd/vGPUmonitor/main.go:11-30 — validate HOOK_PATH, serve
Prometheus (:9394), run the watch/feedback loop (5 s).  The host-pid
mapper replaces the reference's cgroup+NVML scan (feedback.go:83-162):
with hostPID, /proc/<host>/status NSpid's last entry is the container-ns
pid, which is what the interceptor wrote into its slot.
"""
from __future__ import annotations

import argparse
import glob
import logging
import os
import re
import time
from typing import Dict, Optional

from ..plugin.kfd import enumerate_gpus
from ..utils.kubeclient import KubeClient, RestKubeClient
from .feedback import FeedbackLoop
from .metrics import MonitorCollector, serve_metrics
from .pathmon import PathMonitor

log = logging.getLogger(__name__)


def map_host_pids(proc_root: str = "/proc") -> Dict[int, int]:
    """container-ns pid -> host pid, from NSpid (requires hostPID)."""
    mapping: Dict[int, int] = {}
    for status in glob.glob(os.path.join(proc_root, "[0-9]*", "status")):
        try:
            with open(status) as f:
                for line in f:
                    if line.startswith("NSpid:"):
                        parts = line.split()[1:]
                        if len(parts) >= 2:
                            mapping[int(parts[-1])] = int(parts[0])
                        break
        except (OSError, ValueError):
            continue
    return mapping


def update_host_pids(pathmon: PathMonitor, proc_root: str = "/proc") -> None:
    ns_map = map_host_pids(proc_root)
    if not ns_map:
        return
    for entry in pathmon.live_regions():
        snap = entry.region.snapshot()
        L = entry.region.layout
        for slot in range(L["_max_procs"]):
            base = L["procs"] + slot * L["_proc_slot_size"]
            pid = entry.region._i32(base + L["_proc_pid"])
            if pid > 0 and pid in ns_map:
                entry.region.set_host_pid(slot, ns_map[pid])


def main(argv=None):
    p = argparse.ArgumentParser("vgpu-monitor")
    p.add_argument("--hook-path", default=os.environ.get("HOOK_PATH", "/usr/local/vgpu"))
    p.add_argument("--metrics-port", type=int, default=9394)
    p.add_argument("--grpc-bind", default="0.0.0.0:9395",
                   help="NodeVGPUInfo gRPC bind ('' disables)")
    p.add_argument("--interval", type=float, default=5.0)
    p.add_argument("--soft-cores", action="store_true",
                   default=os.environ.get("VGPU_MONITOR_SOFT_CORES", "") == "1")
    args = p.parse_args(argv)
    logging.basicConfig(level=logging.INFO)

    if not args.hook_path:
        raise SystemExit("HOOK_PATH is required")

    gpus = enumerate_gpus()
    pathmon = PathMonitor(args.hook_path)

    busy_paths = {g.uuid: f"/sys/class/drm/card{g.drm_card}/device/gpu_busy_percent"
                  for g in gpus}

    def busy_reader(uuid: str):
        path = busy_paths.get(uuid)
        if path is None:
            return -1
        try:
            with open(path) as f:
                return int(f.read().strip())
        except (OSError, ValueError):
            return -1

    feedback = FeedbackLoop(pathmon, soft_cores=args.soft_cores,
                            busy_reader=busy_reader if gpus else None,
                            interval_s=args.interval)
    collector = MonitorCollector(pathmon, gpus)
    serve_metrics(collector, args.metrics_port)
    if args.grpc_bind:
        from . import noderpc

        noderpc.serve(pathmon, node_name=os.environ.get("NODE_NAME", ""),
                      bind=args.grpc_bind)
    log.info("monitor up: %d GPUs, metrics :%d, grpc %s", len(gpus),
             args.metrics_port, args.grpc_bind or "off")

    client: Optional[KubeClient] = None
    try:
        client = RestKubeClient()
    except Exception as e:
        log.warning("no k8s API access (%s); GC by dir mtime only", e)

    while True:
        live_uids = set()
        if client is not None:
            try:
                live_uids = {p.uid for p in client.list_pods()}
            except Exception as e:
                log.warning("pod list failed: %s", e)
                live_uids = {e.pod_uid for e in pathmon.entries.values()}
        else:
            live_uids = {e.pod_uid for e in pathmon.entries.values()}
        try:
            pathmon.scan(live_uids)
            update_host_pids(pathmon)
            feedback.observe_once()
        except Exception:
            log.exception("monitor tick failed")  # keep the loop alive
        time.sleep(args.interval)


if __name__ == "__main__":
    main()
