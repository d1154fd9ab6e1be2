#!/usr/bin/env python3
"""Oversubscription benchmark (BASELINE config 5: virtual device memory).

Under LD_PRELOAD with VGPU_OVERSUBSCRIBE=true every device allocation
becomes hipMallocManaged; with HSA_XNACK=1 pages migrate between HBM and
host DRAM on demand.  This measures sequential touch bandwidth over a
working set LARGER than the quota — i.e., the cost of paging — vs a
fits-in-HBM control.

The working set is bounded (default 1.15x of quota, quota = physical HBM)
so host-RAM spill stays ~40 GB: safe for the box.

Run (GPU box):
  VGPU_DEVICE_MEMORY_SHARED_CACHE=/tmp/o.cache \
  VGPU_DEVICE_MEMORY_LIMIT=400g VGPU_OVERSUBSCRIBE=true HSA_XNACK=1 \
  LD_PRELOAD=.../libvgpu-hip.so python benchmarks/oversub_bench.py --ratio 1.15
"""
import argparse
import json
import time

import torch


def touch_pass(chunks):
    for c in chunks:
        c.add_(1.0)
    torch.cuda.synchronize()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--ratio", type=float, default=1.15,
                   help="working set as a multiple of physical HBM")
    p.add_argument("--chunk-gb", type=float, default=4.0)
    p.add_argument("--passes", type=int, default=3)
    p.add_argument("--target-gb", type=float, default=0.0,
                   help="absolute working set; use this under the preload, "
                        "where total_memory reports the (larger) QUOTA and "
                        "ratio-of-total would blow past the quota itself")
    args = p.parse_args()

    assert torch.cuda.is_available()
    hbm = torch.cuda.get_device_properties(0).total_memory
    target = int(args.target_gb * (1 << 30)) if args.target_gb \
        else int(hbm * args.ratio)
    # never exceed the allocator's own ceiling (quota) — leave 5% headroom
    target = min(target, int(hbm * 0.95))
    chunk = int(args.chunk_gb * (1 << 30))
    n_chunks = max(1, target // chunk)

    # bulk-migrate each chunk into HBM with hipMemPrefetchAsync: XNACK
    # demand faults back fresh pages one retry at a time (minutes per GB),
    # while prefetch DMA-moves whole ranges at engine speed — the
    # MI355X-native way to establish residency
    import ctypes
    hip = ctypes.CDLL("libamdhip64.so")
    hip.hipMemPrefetchAsync.argtypes = [
        ctypes.c_void_p, ctypes.c_size_t, ctypes.c_int, ctypes.c_void_p]

    chunks = []
    t0 = time.perf_counter()
    for i in range(n_chunks):
        x = torch.empty(chunk // 4, dtype=torch.float32, device="cuda")
        hip.hipMemPrefetchAsync(x.data_ptr(), chunk, 0, None)
        torch.cuda.synchronize()
        chunks.append(x)
    torch.cuda.synchronize()
    alloc_s = time.perf_counter() - t0

    # first pass faults everything in (and pages the tail to host)
    t0 = time.perf_counter()
    touch_pass(chunks)
    first_s = time.perf_counter() - t0

    t0 = time.perf_counter()
    for _ in range(args.passes):
        touch_pass(chunks)
    steady_s = (time.perf_counter() - t0) / args.passes

    total_gb = n_chunks * chunk / (1 << 30)
    print(json.dumps({
        "metric": "oversubscribed touch bandwidth",
        "working_set_gb": round(total_gb, 1),
        "hbm_gb": round(hbm / (1 << 30), 1),
        "ratio": round(total_gb / (hbm / (1 << 30)), 3),
        "alloc_s": round(alloc_s, 2),
        "first_touch_gbps": round(total_gb / first_s, 1),
        "steady_gbps": round(total_gb / steady_s, 1),
        "passes": args.passes,
    }))


if __name__ == "__main__":
    main()
