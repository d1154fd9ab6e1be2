#!/usr/bin/env python3
"""Density benchmark: N co-located "pods" on one MI355X (BASELINE config 3).

Simulates the published density claim (10 inference pods per GPU,
reference README.md:40) without a cluster: N worker processes, each under
LD_PRELOAD libvgpu-hip.so with

  - HBM quota        = quota_pct% of the card (default 100/N, like
                       gpumem-percentage),
  - CU soft limit    = 100/N percent (VGPU_DEVICE_CU_LIMIT),
  - its own shared region (one per "container", as the plugin injects),

each running ResNet-V2-50 inference (b50 @346^2, the ai-benchmark case)
for --seconds wall-clock.  Reports per-worker and aggregate samples/s plus
the fairness spread (max/min) — the number the reference only publishes as
a chart.

Usage: python benchmarks/density_bench.py --pods 10 --seconds 30
"""
import argparse
import json
import os
import subprocess
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
LIBVGPU = REPO / "k8s_device_plugin_amd" / "csrc" / "libvgpu-hip.so"


def worker(args):
    import torch

    sys.path.insert(0, str(REPO))
    from k8s_device_plugin_amd.models import zoo

    case = zoo.CASES[args.case]
    dev = torch.device("cuda", 0)
    model = zoo.build(case, dev)
    batch = zoo.synthetic_batch(case, dev)
    for _ in range(3):
        zoo.step(case, model, batch, None)
    torch.cuda.synchronize()
    print("READY", flush=True)
    assert sys.stdin.readline().strip() == "GO"
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < args.settle:
        zoo.step(case, model, batch, None)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    steps = 0
    while time.perf_counter() - t0 < args.seconds:
        zoo.step(case, model, batch, None)
        steps += 1
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({"samples_per_s": case.batch * steps / dt,
                      "steps": steps}), flush=True)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--pods", type=int, default=10)
    p.add_argument("--seconds", type=float, default=30.0)
    p.add_argument("--settle", type=float, default=6.0,
                   help="untimed run-in before the timed window (arbiter "
                        "steady state)")
    p.add_argument("--case", default="resnet50_inf")
    p.add_argument("--quota-pct", type=int, default=0,
                   help="HBM percent per pod (default 100/pods)")
    p.add_argument("--cu-pct", type=int, default=0,
                   help="CU percent per pod (default 100/pods)")
    p.add_argument("--no-cu-mask", action="store_true",
                   help="skip the hard HSA_CU_MASK partition (soft limiter only)")
    p.add_argument("--arbitrate", action="store_true",
                   help="run the monitor's fair-throttle arbitration loop "
                        "over the pods' regions (as the vGPU monitor does "
                        "in production)")
    p.add_argument("--prewarm", action="store_true",
                   help="run one unmasked pass first to populate the MIOpen "
                        "find-db (keyed by conv config, not CU mask) so the "
                        "masked pods skip autotuning")
    p.add_argument("--worker", action="store_true", help=argparse.SUPPRESS)
    args = p.parse_args()
    if args.worker:
        worker(args)
        return

    import torch

    total_mem = torch.cuda.get_device_properties(0).total_memory \
        if torch.cuda.is_available() else 288 << 30
    quota_pct = args.quota_pct or max(1, 100 // args.pods)
    cu_pct = args.cu_pct or max(1, 100 // args.pods)
    quota_mib = total_mem * quota_pct // 100 // (1 << 20)

    if args.prewarm:
        import torch as _t

        sys.path.insert(0, str(REPO))
        from k8s_device_plugin_amd.models import zoo as _zoo

        _case = _zoo.CASES[args.case]
        _model = _zoo.build(_case, _t.device("cuda", 0))
        _batch = _zoo.synthetic_batch(_case, _t.device("cuda", 0))
        for _ in range(3):
            _zoo.step(_case, _model, _batch, None)
        _t.cuda.synchronize()
        del _model, _batch
        _t.cuda.empty_cache()
        print("prewarm done", flush=True)

    # hard partition: disjoint CU masks, exactly as the plugin injects at
    # Allocate (ops/cumask.py) — 10 pods on 256 CUs -> 25 CUs each.  The
    # soft token-bucket limiter stays on top (both layers, the production
    # stack).  Fairness comes from the hardware partition.
    total_cus = 256
    chunk = total_cus // args.pods
    masks = []
    for i in range(args.pods):
        lo = i * chunk
        hi = lo + chunk - 1
        masks.append(f"0:{lo}-{hi}")

    procs = []
    caches = []
    for i in range(args.pods):
        cache = tempfile.NamedTemporaryFile(prefix=f"density-{i}-",
                                            suffix=".cache", delete=False)
        cache.close()
        caches.append(cache.name)
        env = dict(os.environ)
        env.update({
            "LD_PRELOAD": str(LIBVGPU),
            "VGPU_DEVICE_MEMORY_LIMIT": f"{quota_mib}m",
            "VGPU_DEVICE_CU_LIMIT": str(cu_pct),
            "VGPU_DEVICE_MEMORY_SHARED_CACHE": cache.name,
            "VGPU_DEVICE_UUIDS": f"GPU-density-{i}",
        })
        if not args.no_cu_mask:
            env["HSA_CU_MASK"] = masks[i]
        procs.append(subprocess.Popen(
            [sys.executable, __file__, "--worker", "--case", args.case,
             "--seconds", str(args.seconds), "--settle", str(args.settle)],
            env=env, stdin=subprocess.PIPE, stdout=subprocess.PIPE, text=True))

    arb_stop = None
    if args.arbitrate:
        import glob
        import threading

        from k8s_device_plugin_amd.monitor.region import SharedRegion

        def find_busy_path():
            for path in sorted(glob.glob(
                    "/sys/class/drm/card*/device/gpu_busy_percent")):
                return path
            return None

        busy_path = find_busy_path()
        arb_stop = threading.Event()

        def arbitrate():
            # token-bound median controller (monitor/feedback.py
            # _arbitrate): tighten while fewer than half the launching
            # pods sample token-bound, relax slowly otherwise — fairness
            # from the buckets alone, no host busy% (which reads the
            # wrong card on multi-DRM hosts and unthrottles everyone)
            from k8s_device_plugin_amd.monitor.arbiter import ScaleArbiter

            regions = {}
            prev_tokens = {}
            arb = ScaleArbiter()
            scale = 1.0
            while not arb_stop.is_set():
                # late-attach: the workers create/initialize their regions
                # after launch
                for c in caches:
                    if c in regions:
                        continue
                    try:
                        r = SharedRegion(c)
                        if r.valid:
                            regions[c] = r
                    except (OSError, ValueError):
                        pass
                active = 0
                bound = 0
                for c, r in regions.items():
                    try:
                        tokens = r.get_core_tokens(0)
                        fill = r.get_token_fill_rate(0)
                    except (OSError, ValueError):
                        continue
                    cap = max(1.0, fill * 0.25)
                    moved = prev_tokens.get(c) != tokens
                    prev_tokens[c] = tokens
                    if tokens <= 0:
                        active += 1
                        bound += 1
                    elif moved or tokens < 0.9 * cap:
                        active += 1
                scale = arb.tick(active, bound)
                now = time.monotonic_ns()
                for r in regions.values():
                    try:
                        r.set_monitor_interval(0.25)
                        r.set_monitor_scale(0, scale, now)
                    except (OSError, ValueError):
                        pass
                arb_stop.wait(0.25)

        threading.Thread(target=arbitrate, daemon=True).start()

    # barrier: wait until every pod is warm, then release together
    for pr in procs:
        assert pr.stdout.readline().strip() == "READY"
    for pr in procs:
        pr.stdin.write("GO\n")
        pr.stdin.flush()

    results = []
    for pr in procs:
        line = ""
        for line in pr.stdout:
            if line.startswith("{"):
                break
        pr.wait(timeout=120)
        results.append(json.loads(line))
    if arb_stop is not None:
        arb_stop.set()
    for c in caches:
        os.unlink(c)

    rates = [r["samples_per_s"] for r in results]
    print(json.dumps({
        "metric": "co-located pods/GPU at equal HBM+CU quota",
        "value": args.pods,
        "unit": "pods",
        "aggregate_samples_per_s": round(sum(rates), 2),
        "per_pod_samples_per_s": [round(r, 2) for r in rates],
        "fairness_max_over_min": round(max(rates) / max(min(rates), 1e-9), 3),
        "quota_pct": quota_pct,
        "cu_pct": cu_pct,
        "hard_cu_mask": not args.no_cu_mask,
        "arbitrated": bool(args.arbitrate),
        "case": args.case,
        "seconds": args.seconds,
    }))


if __name__ == "__main__":
    main()
