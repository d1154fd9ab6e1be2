#!/bin/bash
# Round-2 GPU call 1: gpu pytest + MIOpen find-db capture (full 10-case
# suite, cold) + amd-smi preload-crash triage (ROUND2_NOTES item 5).
set -x
cd /root/repo
mkdir -p gpurun_out/miopen_udb gpurun_out/miopen_cache

# 1. GPU test suite with the round-2 library
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/gputest_r2a.log 2>&1
echo "pytest rc=$?"
tail -5 gpurun_out/gputest_r2a.log

# 2. Full 10-case bench, cold: pays the MIOpen autotune once and captures
#    the user find-db + compiled-kernel cache for shipping in-tree.
export MIOPEN_USER_DB_PATH=/root/repo/gpurun_out/miopen_udb
export MIOPEN_CUSTOM_CACHE_DIR=/root/repo/gpurun_out/miopen_cache
timeout 1500 python bench.py --cases all --steps 2 --warmup 1 > gpurun_out/bench_all_cold.log 2>&1
echo "bench rc=$?"
tail -3 gpurun_out/bench_all_cold.log
du -sh gpurun_out/miopen_udb gpurun_out/miopen_cache

# 3. amd-smi PLT-link triage matrix
cd k8s_device_plugin_amd/csrc
export LD_LIBRARY_PATH=/opt/rocm/lib
{
  echo "=== bare (control) ==="
  timeout 60 ./test/amdsmi_consumer; echo "rc=$?"
  echo "=== empty preload ==="
  timeout 60 env LD_PRELOAD=$PWD/test/empty_preload.so ./test/amdsmi_consumer; echo "rc=$?"
  echo "=== full libvgpu, control disabled, BT ==="
  timeout 60 env AMDSMI_BT=1 VGPU_DISABLE_CONTROL=1 LD_PRELOAD=$PWD/libvgpu-hip.so ./test/amdsmi_consumer; echo "rc=$?"
  echo "=== nodlsym variant ==="
  timeout 60 env AMDSMI_BT=1 VGPU_DISABLE_CONTROL=1 LD_PRELOAD=$PWD/libvgpu-hip-nodlsym.so ./test/amdsmi_consumer; echo "rc=$?"
  echo "=== nodlopen+nodlsym variant ==="
  timeout 60 env AMDSMI_BT=1 VGPU_DISABLE_CONTROL=1 LD_PRELOAD=$PWD/libvgpu-hip-nodlopen.so ./test/amdsmi_consumer; echo "rc=$?"
  echo "=== full libvgpu, control ENABLED (quota view) ==="
  timeout 60 env AMDSMI_BT=1 VGPU_DEVICE_MEMORY_SHARED_CACHE=/tmp/amdsmi-triage.cache VGPU_DEVICE_MEMORY_LIMIT=73728m LD_PRELOAD=$PWD/libvgpu-hip.so ./test/amdsmi_consumer; echo "rc=$?"
  echo "=== full libvgpu, LD_DEBUG=libs (tail) ==="
  timeout 60 env LD_DEBUG=libs VGPU_DISABLE_CONTROL=1 LD_PRELOAD=$PWD/libvgpu-hip.so ./test/amdsmi_consumer 2>&1 | tail -40
} > /root/repo/gpurun_out/amdsmi_triage_r2.log 2>&1
echo "triage done"
tail -60 /root/repo/gpurun_out/amdsmi_triage_r2.log
