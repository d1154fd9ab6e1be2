#!/usr/bin/env python3
"""Flagship benchmark: vGPU overhead vs native on ai-benchmark (MI355X).

Measures the reference's headline metric (BASELINE.json): aggregate
ai-benchmark throughput for the SAME workload run (a) bare ("native device
plugin") and (b) under the vGPU enforcement stack — LD_PRELOAD
libvgpu-hip.so with a 50% HBM quota, the published job shape
(/root/reference/benchmarks/ai-benchmark/Hami/ai-benchmark.yml:14-19:
gpumem-percentage: 50).  Reported value = overhead percent (lower better).

Per rank (one GPU, launched by torch.distributed.run for N>1):
  1. probe the GPU's HBM size (subprocess, no preload);
  2. native phase: worker subprocess builds the cases, warms up W steps,
     signals READY; parent barriers all ranks (gloo), sends GO; worker times
     EXACTLY K steps with torch.cuda.synchronize around the timed region;
  3. vgpu phase: identical worker, plus LD_PRELOAD + VGPU_DEVICE_MEMORY_LIMIT
     = quota% of HBM + a fresh shared region;
  4. all-reduce: aggregate throughputs (sum) and step time (max) over ranks.

A "step" = one iteration of every selected ai-benchmark case (default:
ResNet-V2-50 inference b50@346^2 + training b20@346^2; --cases all runs the
full 10-case suite).  Synthetic data, random-init weights, fp32 (the
reference suite's dtype).
"""
import argparse
import json
import os
import subprocess
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent
LIBVGPU = REPO / "k8s_device_plugin_amd" / "csrc" / "libvgpu-hip.so"


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--cases", default="resnet50_inf,resnet50_train",
                   help="comma list or 'all'")
    p.add_argument("--quota-pct", type=int, default=50)
    p.add_argument("--worker", action="store_true", help=argparse.SUPPRESS)
    p.add_argument("--device", type=int, default=0, help=argparse.SUPPRESS)
    return p.parse_args()


def resolve_cases(spec):
    from k8s_device_plugin_amd.models import zoo

    if spec == "all":
        return list(zoo.CASES.keys())
    return [c.strip() for c in spec.split(",") if c.strip()]


# ---------------------------------------------------------------------------
# Worker: runs inside the (optionally preloaded) subprocess.
# ---------------------------------------------------------------------------
def worker_main(args):
    import torch

    from k8s_device_plugin_amd.models import zoo

    assert torch.cuda.is_available(), "worker needs a GPU"
    dev = torch.device("cuda", 0)
    torch.backends.cudnn.benchmark = True
    names = resolve_cases(args.cases)
    setups = []
    for name in names:
        case = zoo.CASES[name]
        model = zoo.build(case, dev)
        batch = zoo.synthetic_batch(case, dev)
        opt = (torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
               if case.phase == "training" else None)
        setups.append((case, model, batch, opt))

    for _ in range(args.warmup):
        for case, model, batch, opt in setups:
            zoo.step(case, model, batch, opt)
    torch.cuda.synchronize()

    print("READY", flush=True)
    line = sys.stdin.readline()
    assert line.strip() == "GO", f"unexpected control message: {line!r}"

    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        for case, model, batch, opt in setups:
            zoo.step(case, model, batch, opt)
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    samples_per_step = sum(c.batch for c, _, _, _ in setups)
    print(json.dumps({
        "elapsed_s": elapsed,
        "steps": args.steps,
        "samples_per_step": samples_per_step,
        "samples_per_s": samples_per_step * args.steps / elapsed,
    }), flush=True)


# ---------------------------------------------------------------------------
# Parent: per-rank orchestration.
# ---------------------------------------------------------------------------
def probe_total_mem(device_env):
    code = ("import torch;"
            "print(torch.cuda.get_device_properties(0).total_memory)")
    out = subprocess.run([sys.executable, "-c", code], env=device_env,
                         capture_output=True, text=True, timeout=600)
    if out.returncode != 0:
        raise RuntimeError(f"GPU probe failed: {out.stderr[-800:]}")
    return int(out.stdout.strip().splitlines()[-1])


def run_phase(args, device_env, barrier):
    cmd = [sys.executable, str(REPO / "bench.py"), "--worker",
           "--cases", args.cases, "--steps", str(args.steps),
           "--warmup", str(args.warmup)]
    proc = subprocess.Popen(cmd, env=device_env, stdin=subprocess.PIPE,
                            stdout=subprocess.PIPE, text=True, cwd=str(REPO))
    result = None
    try:
        for line in proc.stdout:
            if line.strip() == "READY":
                barrier()  # all ranks' workers are warm
                proc.stdin.write("GO\n")
                proc.stdin.flush()
            elif line.startswith("{"):
                result = json.loads(line)
        proc.wait(timeout=60)
    finally:
        if proc.poll() is None:
            proc.kill()
    barrier()  # all ranks finished the timed region
    if result is None:
        raise RuntimeError("worker produced no result")
    return result


def main():
    args = parse_args()
    if args.worker:
        worker_main(args)
        return

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    dist = None
    if world > 1:
        import torch.distributed as td

        td.init_process_group(backend="gloo")
        dist = td

    def barrier():
        if dist is not None:
            dist.barrier()

    base_env = dict(os.environ)
    # one GPU per rank; the worker sees it as cuda:0
    base_env["CUDA_VISIBLE_DEVICES"] = str(local_rank)
    base_env["HIP_VISIBLE_DEVICES"] = str(local_rank)
    base_env.pop("LD_PRELOAD", None)
    # keep workers out of the parent's rendezvous
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR", "MASTER_PORT",
              "GROUP_RANK", "LOCAL_WORLD_SIZE", "TORCHELASTIC_RUN_ID"):
        base_env.pop(k, None)

    total_mem = probe_total_mem(base_env)
    quota_bytes = total_mem * args.quota_pct // 100
    quota_mib = quota_bytes // (1 << 20)

    t_job0 = time.perf_counter()
    native = run_phase(args, base_env, barrier)

    vgpu_env = dict(base_env)
    cache = tempfile.NamedTemporaryFile(prefix="vgpu-bench-", suffix=".cache",
                                        delete=False)
    cache.close()
    vgpu_env.update({
        "LD_PRELOAD": str(LIBVGPU),
        "VGPU_DEVICE_MEMORY_LIMIT": f"{quota_mib}m",
        "VGPU_DEVICE_MEMORY_SHARED_CACHE": cache.name,
        "VGPU_DEVICE_UUIDS": f"GPU-bench-{local_rank}",
    })
    if not LIBVGPU.exists():
        raise RuntimeError(f"{LIBVGPU} missing — run __graft_entry__.build()")
    vgpu = run_phase(args, vgpu_env, barrier)
    os.unlink(cache.name)
    wall = time.perf_counter() - t_job0

    import torch

    def agg(val, op):
        if dist is None:
            return val
        t = torch.tensor([val], dtype=torch.float64)
        dist.all_reduce(t, op=op)
        return float(t.item())

    import torch.distributed as td_ops
    SUM = td_ops.ReduceOp.SUM if dist else None
    MAX = td_ops.ReduceOp.MAX if dist else None

    native_total = agg(native["samples_per_s"], SUM)
    vgpu_total = agg(vgpu["samples_per_s"], SUM)
    vgpu_ms_per_step = agg(vgpu["elapsed_s"] * 1000.0 / args.steps, MAX)
    overhead_pct = (native_total - vgpu_total) / native_total * 100.0

    if rank == 0:
        print(json.dumps({
            "metric": "vGPU overhead vs native (%) on ai-benchmark",
            "value": round(overhead_pct, 3),
            "unit": "percent",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(vgpu_ms_per_step, 3),
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "ai-benchmark:" + args.cases,
                "global_batch": sum(
                    __import__("k8s_device_plugin_amd.models.zoo",
                               fromlist=["CASES"]).CASES[c].batch
                    for c in resolve_cases(args.cases)) * world,
                "seq_len": None,
                "parallelism": f"pods-per-gpu quota={args.quota_pct}%mem",
                "quota_mib": quota_mib,
                "native_samples_per_s": round(native_total, 2),
                "vgpu_samples_per_s": round(vgpu_total, 2),
                "enforcement": "LD_PRELOAD libvgpu-hip.so, hard HBM cap",
            },
        }), flush=True)
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
