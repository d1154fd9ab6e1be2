"""amd-smi (libamd_smi) quota spoofing against the fake library.

The reference spoofs nvidia-smi via ~260 nvml hooks;
rocm-smi is covered by smi.c and amd-smi — the modern CLI — by amdsmi.c.
"""
import json
import os
import subprocess
from pathlib import Path

import pytest

CSRC = Path(__file__).resolve().parent.parent / "k8s_device_plugin_amd" / "csrc"
LIBVGPU = CSRC / "libvgpu-hip.so"
FAKEDIR = CSRC / "fakeamdsmi"
CONSUMER = CSRC / "test" / "amdsmi_consumer"

GIB = 1 << 30
MIB = 1 << 20


@pytest.fixture(scope="session", autouse=True)
def build_native():
    if not (LIBVGPU.exists() and CONSUMER.exists()):
        subprocess.run(["make"], cwd=CSRC, check=True, capture_output=True)


def run_consumer(cache, limit=None, preload=True, extra=None):
    env = dict(os.environ)
    env["LD_LIBRARY_PATH"] = str(FAKEDIR)
    if preload:
        env["LD_PRELOAD"] = str(LIBVGPU)
    env["VGPU_DEVICE_MEMORY_SHARED_CACHE"] = str(cache)
    env["VGPU_REAL_AMDSMI_PATH"] = str(FAKEDIR / "libamd_smi.so")
    env["VGPU_REAL_HIP_PATH"] = str(CSRC / "fakehip" / "libamdhip64.so")
    if limit:
        env["VGPU_DEVICE_MEMORY_LIMIT"] = limit
    env.update(extra or {})
    out = subprocess.run([str(CONSUMER)], env=env, capture_output=True,
                         text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    return [json.loads(l) for l in out.stdout.splitlines() if l.startswith("{")]


def test_amdsmi_reports_quota(tmp_path):
    devs = run_consumer(tmp_path / "r.cache", limit="73728m")
    assert len(devs) == 2
    # device 0 is the container's vGPU: totals clamp to the 72 GiB quota
    assert devs[0]["total"] == 72 * GIB
    assert devs[0]["used"] == 0          # nothing allocated by this container
    assert devs[0]["vram_total_mb"] == 73728
    assert devs[0]["vram_used_mb"] == 0


def test_amdsmi_passthrough_without_preload(tmp_path):
    devs = run_consumer(tmp_path / "r.cache", limit="73728m", preload=False)
    assert devs[0]["total"] == 288 * GIB
    assert devs[0]["used"] == 200 * GIB


def test_amdsmi_disable_control(tmp_path):
    devs = run_consumer(tmp_path / "r.cache", limit="73728m",
                        extra={"VGPU_DISABLE_CONTROL": "1"})
    assert devs[0]["total"] == 288 * GIB


def test_amdsmi_activity_clamped_to_cu_limit(tmp_path):
    devs = run_consumer(tmp_path / "r.cache", limit="73728m",
                        extra={"VGPU_DEVICE_CU_LIMIT": "25"})
    assert devs[0]["gfx"] == 25   # physical 90 clamped to quota
    devs = run_consumer(tmp_path / "r2.cache", limit="73728m")
    assert devs[0]["gfx"] == 90   # no CU limit -> passthrough


def test_embedded_rsmi_resolved_via_rtld_next(tmp_path):
    """ODR-safety regression test for the round-1 PLT-link SIGBUS: a
    libamd_smi consumer's rsmi_* call is interposed by the hook, which
    must resolve the REAL implementation from the in-process libamd_smi
    (RTLD_NEXT — the fake's embedded-rsmi counter proves it) instead of
    dlopening a second, clashing librocm_smi64."""
    env_extra = {"AMDSMI_CALL_RSMI": "1"}
    rows = run_consumer(tmp_path / "r.cache", limit="73728m",
                        extra=env_extra)
    rsmi_rows = [r for r in rows if "rsmi_total" in r]
    assert rsmi_rows, rows
    r = rsmi_rows[0]
    assert r["rsmi_rc"] == 0
    assert r["rsmi_total"] == 73728 * MIB      # quota view applied
    assert r["embedded_calls"] >= 1            # served by the embedded copy
