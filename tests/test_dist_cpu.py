"""Multi-process distributed-path test (gloo, world_size 2, CPU-only).

bench.py coordinates one rank per GPU with torch.distributed (gloo
rendezvous for barriers + SUM/MAX aggregation over ranks).  This exercises
that exact aggregation contract here on CPU so the N>1 path is correct by
construction before the driver's 8-GPU round-end run.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _rank_main(rank, world, port, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(port),
        "RANK": str(rank),
        "WORLD_SIZE": str(world),
    })
    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    try:
        # the bench.py aggregation: SUM of throughputs, MAX of step time
        thpt = torch.tensor([1000.0 * (rank + 1)], dtype=torch.float64)
        dist.all_reduce(thpt, op=dist.ReduceOp.SUM)
        step_ms = torch.tensor([10.0 + 5 * rank], dtype=torch.float64)
        dist.all_reduce(step_ms, op=dist.ReduceOp.MAX)
        dist.barrier()
        q.put((rank, float(thpt.item()), float(step_ms.item())))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_gloo_world2_sum_and_max():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29517
    procs = [ctx.Process(target=_rank_main, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, total, step = q.get(timeout=90)
        results[rank] = (total, step)
    for p in procs:
        p.join(timeout=30)
        assert p.exitcode == 0
    # every rank sees the same aggregates: 1000+2000 and max(10,15)
    assert results[0] == (3000.0, 15.0)
    assert results[1] == (3000.0, 15.0)
