#!/bin/bash
# Round-2 GPU call 2: full gpu pytest (CPX isolation, beyond-physical
# oversubscription, amd-smi ODR fix) + the default bench (10 cases, warm
# MIOpen, colocated + density phases) + oversubscription bench record.
set -x
cd /root/repo

# 1. GPU test suite (now 14 tests incl. CPX + oversub + amdsmi PLT)
timeout 1400 python -m pytest tests -m gpu -q > gpurun_out/gputest_r2b.log 2>&1
echo "pytest rc=$?"
tail -8 gpurun_out/gputest_r2b.log

# 2. Default bench exactly as the driver will run it
timeout 1200 python bench.py > gpurun_out/bench_default_r2.log 2>&1
echo "bench rc=$?"
tail -2 gpurun_out/bench_default_r2.log

# 3. Oversubscription bench (BASELINE config 5 record)
timeout 600 env \
  VGPU_DEVICE_MEMORY_SHARED_CACHE=/tmp/oversub.cache \
  VGPU_DEVICE_MEMORY_LIMIT=409600m VGPU_OVERSUBSCRIBE=true HSA_XNACK=1 \
  LD_PRELOAD=/root/repo/k8s_device_plugin_amd/csrc/libvgpu-hip.so \
  python benchmarks/oversub_bench.py --ratio 1.1 --passes 2 \
  > gpurun_out/oversub_r2.log 2>&1
echo "oversub rc=$?"
tail -2 gpurun_out/oversub_r2.log
