"""vgpu-top: node-local view of every container's vGPU enforcement state.

Reads the live shared regions under <hook>/containers (the same source the
monitor exports to Prometheus) and prints one row per container: quota,
usage, CU limit, priority, block state.  `--watch` refreshes.

Run on a node (or in the monitor container):
  python -m k8s_device_plugin_amd.tools.vgpu_top [--hook-path /usr/local/vgpu]
"""
from __future__ import annotations

import argparse
import os
import time
from typing import List

from ..monitor.pathmon import PathMonitor

GIB = 1 << 30


def rows(pathmon: PathMonitor) -> List[dict]:
    out = []
    for e in pathmon.live_regions():
        try:
            snap = e.region.snapshot()
        except Exception:
            continue
        for dev in range(snap.num_devices or 1):
            uuid = snap.uuids[dev] if dev < len(snap.uuids) else ""
            limit = snap.limit[dev] if dev < len(snap.limit) else 0
            used = snap.device_usage(dev) if snap.procs else 0
            out.append({
                "pod": e.pod_uid,
                "ctr": e.container,
                "dev": dev,
                "uuid": uuid,
                "used_gib": used / GIB,
                "limit_gib": limit / GIB,
                "cu_pct": snap.sm_limit[dev] if dev < len(snap.sm_limit) else 0,
                "procs": len(snap.procs),
                "prio": snap.priority,
                "state": ("BLOCKED" if snap.recent_kernel < 0 else
                          "active" if snap.recent_kernel > 0 else "idle"),
                "oversub": bool(snap.oversubscribe),
                "scale": e.region.get_monitor_scale(dev),
            })
    return out


def render(rs: List[dict]) -> str:
    hdr = (f"{'POD':<38} {'CTR':<12} {'DEV':>3} {'USED':>9} {'LIMIT':>9} "
           f"{'CU%':>4} {'SCALE':>6} {'PROCS':>5} {'PRIO':>4} {'STATE':<8} OS")
    lines = [hdr, "-" * len(hdr)]
    for r in rs:
        lines.append(
            f"{r['pod']:<38} {r['ctr']:<12} {r['dev']:>3} "
            f"{r['used_gib']:>8.1f}G {r['limit_gib']:>8.1f}G "
            f"{r['cu_pct']:>4} {r['scale']:>6.2f} {r['procs']:>5} "
            f"{r['prio']:>4} {r['state']:<8} {'y' if r['oversub'] else '-'}")
    if not rs:
        lines.append("(no live vGPU containers)")
    return "\n".join(lines)


def main(argv=None) -> int:
    p = argparse.ArgumentParser("vgpu-top")
    p.add_argument("--hook-path",
                   default=os.environ.get("HOOK_PATH", "/usr/local/vgpu"))
    p.add_argument("--watch", type=float, default=0.0,
                   help="refresh every N seconds (0 = print once)")
    args = p.parse_args(argv)
    pathmon = PathMonitor(args.hook_path)
    while True:
        # GC decisions belong to the monitor; here every dir is "live"
        pathmon.scan({e.split("_")[0] for e in
                      os.listdir(os.path.join(args.hook_path, "containers"))
                      } if os.path.isdir(
                          os.path.join(args.hook_path, "containers")) else set())
        print(render(rows(pathmon)), flush=True)
        if args.watch <= 0:
            return 0
        time.sleep(args.watch)
        print()


if __name__ == "__main__":
    raise SystemExit(main())
