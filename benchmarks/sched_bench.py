#!/usr/bin/env python3
"""Scheduler control-plane benchmark (CPU-only, no cluster).

BASELINE config 4 at benchmark scale: bin-pack --pods fractional-GPU pods
across --gpus MI355X devices on one node through the REAL filter/bind code
paths (in-memory fake API), reporting scheduling throughput and packing
efficiency.  The reference publishes no such number; this bounds the
control-plane cost of the stack.

  python benchmarks/sched_bench.py --pods 32 --gpus 8
"""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from k8s_device_plugin_amd.device.amd import HANDSHAKE_ANNO, REGISTER_ANNO
from k8s_device_plugin_amd.scheduler.core import Scheduler
from k8s_device_plugin_amd.utils.codec import encode_node_devices
from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
from k8s_device_plugin_amd.utils.types import (
    ContainerSpec,
    DeviceInfo,
    NodeInfo,
    PodInfo,
)

MI355X_MEM_MIB = 294912


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--pods", type=int, default=32)
    p.add_argument("--gpus", type=int, default=8)
    p.add_argument("--mem-mib", type=int, default=0,
                   help="per-pod HBM (default: pack to fill = gpus*total/pods)")
    p.add_argument("--cores", type=int, default=0,
                   help="per-pod CU percent (default 100*gpus/pods)")
    args = p.parse_args()

    mem = args.mem_mib or (args.gpus * MI355X_MEM_MIB) // args.pods
    cores = args.cores or max(1, (100 * args.gpus) // args.pods)

    client = FakeKubeClient()
    devices = [DeviceInfo(id=f"GPU-{i:02d}", count=10, devmem=MI355X_MEM_MIB,
                          devcore=100, type="AMD-Instinct-MI355X",
                          numa=i // max(1, args.gpus // 2), health=True,
                          index=i)
               for i in range(args.gpus)]
    client.add_node(NodeInfo(name="n1", annotations={
        HANDSHAKE_ANNO: "Reported 2026-01-01 00:00:00",
        REGISTER_ANNO: encode_node_devices(devices),
    }))
    sched = Scheduler(client)
    sched.register_from_node_annotations_once()

    placed = 0
    t0 = time.perf_counter()
    for i in range(args.pods):
        pod = PodInfo(
            name=f"p{i}", uid=f"uid-{i}",
            containers=[ContainerSpec(name="c", limits={
                "amd.com/gpu": 1,
                "amd.com/gpumem": mem,
                "amd.com/gpucores": cores,
            })],
        )
        client.add_pod(pod)
        fr = sched.filter(pod, ["n1"])
        if fr.node_names:
            br = sched.bind(pod.name, "default", fr.node_names[0])
            if not br.error:
                placed += 1
                # release the bind lock the way Allocate success would
                from k8s_device_plugin_amd.utils import nodelock
                nodelock.release_node_lock(client, "n1")
    dt = time.perf_counter() - t0

    usage, _ = sched.get_nodes_usage(["n1"])
    mems = [d.usedmem for d in usage["n1"].devices]
    print(json.dumps({
        "metric": "scheduler binpack throughput",
        "pods_requested": args.pods,
        "pods_placed": placed,
        "gpus": args.gpus,
        "per_pod_mem_mib": mem,
        "per_pod_cores": cores,
        "pods_per_second": round(args.pods / dt, 1),
        "ms_per_pod": round(dt * 1000 / args.pods, 3),
        "packing_mem_utilization": round(
            sum(mems) / (args.gpus * MI355X_MEM_MIB), 4),
        "per_gpu_mem_mib": mems,
    }))


if __name__ == "__main__":
    main()
