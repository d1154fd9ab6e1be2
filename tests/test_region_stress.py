"""Shared-region concurrency stress: many processes hammering one region.

Validates the robust-mutex ledger under contention (reference analog: the
shrreg semaphore + owner-pid repair) — final usage must be
exactly the survivors' allocations, crashed holders must not wedge the
region, and a killed process's usage must be pruned.
"""
import json
import os
import signal
import subprocess
import time
from pathlib import Path

import pytest

CSRC = Path(__file__).resolve().parent.parent / "k8s_device_plugin_amd" / "csrc"
LIBVGPU = CSRC / "libvgpu-hip.so"
FAKEDIR = CSRC / "fakehip"
CONSUMER = CSRC / "test" / "hip_consumer"
MIB = 1 << 20


def env_for(cache):
    env = dict(os.environ)
    env.update({
        "LD_LIBRARY_PATH": str(FAKEDIR),
        "LD_PRELOAD": str(LIBVGPU),
        "VGPU_DEVICE_MEMORY_SHARED_CACHE": str(cache),
        "VGPU_REAL_HIP_PATH": str(FAKEDIR / "libamdhip64.so"),
        "VGPU_DEVICE_MEMORY_LIMIT": "100000m",
    })
    return env


def test_concurrent_alloc_free_consistency(tmp_path):
    cache = tmp_path / "r.cache"
    # 8 processes, each: 20 rounds of alloc 10M / free
    cmds = []
    for _ in range(20):
        cmds += ["alloc", str(10 * MIB), "free"]
    procs = [subprocess.Popen([str(CONSUMER)] + cmds, env=env_for(cache),
                              stdout=subprocess.PIPE, text=True)
             for _ in range(8)]
    for p in procs:
        assert p.wait(timeout=120) == 0
        for line in p.stdout.read().splitlines():
            assert json.loads(line)["err"] == 0
    # all alloc/free balanced -> a final observer sees zero usage
    out = subprocess.run([str(CONSUMER), "meminfo"], env=env_for(cache),
                         capture_output=True, text=True, timeout=60)
    info = json.loads(out.stdout.splitlines()[0])
    assert info["free"] == 100000 * MIB


def test_killed_process_usage_pruned(tmp_path):
    cache = tmp_path / "r.cache"
    p = subprocess.Popen(
        [str(CONSUMER), "alloc", str(500 * MIB), "sleep", "60000"],
        env=env_for(cache), stdout=subprocess.PIPE, text=True)
    line = json.loads(p.stdout.readline())
    assert line["err"] == 0
    p.kill()
    p.wait(timeout=30)
    # liveness pruning runs inside usage summation: the dead pid's ledger
    # share must vanish for the next process
    out = subprocess.run([str(CONSUMER), "meminfo"], env=env_for(cache),
                         capture_output=True, text=True, timeout=60)
    info = json.loads(out.stdout.splitlines()[0])
    assert info["free"] == 100000 * MIB


def test_region_survives_sigkill_mid_traffic(tmp_path):
    """Kill workers at random points; the region must stay usable (robust
    mutex dead-owner recovery) and converge to zero usage."""
    cache = tmp_path / "r.cache"
    cmds = []
    for _ in range(50):
        cmds += ["alloc", str(5 * MIB), "free"]
    procs = [subprocess.Popen([str(CONSUMER)] + cmds, env=env_for(cache),
                              stdout=subprocess.DEVNULL)
             for _ in range(6)]
    time.sleep(0.2)
    for p in procs[:3]:
        p.send_signal(signal.SIGKILL)
    for p in procs:
        p.wait(timeout=120)
    out = subprocess.run([str(CONSUMER), "meminfo"], env=env_for(cache),
                         capture_output=True, text=True, timeout=60)
    info = json.loads(out.stdout.splitlines()[0])
    assert info["err"] == 0
    assert info["free"] == 100000 * MIB
