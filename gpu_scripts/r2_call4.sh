#!/bin/bash
# Round-2 GPU call 4 (final): density fairness with limiter debug,
# prefetch-based beyond-physical paging test, prefetch-based oversub
# bench record.
set -x
cd /root/repo

# 1. Density fairness + limiter introspection
timeout 420 python bench.py --density-only --density-seconds 20 \
  > gpurun_out/density_r2c.log 2>&1
echo "density rc=$?"
tail -2 gpurun_out/density_r2c.log

# 2. Beyond-physical paging via bulk prefetch
timeout 650 python -m pytest "tests/test_gpu.py::TestOversubscriptionReal::test_beyond_physical_hbm_pages_and_computes" -q -rs \
  > gpurun_out/paging_test_r2b.log 2>&1
echo "paging rc=$?"
tail -6 gpurun_out/paging_test_r2b.log

# 3. Oversubscription touch-bandwidth record (296 GB on 288 GB)
timeout 500 env \
  VGPU_DEVICE_MEMORY_SHARED_CACHE=/tmp/oversub.cache \
  VGPU_DEVICE_MEMORY_LIMIT=409600m VGPU_OVERSUBSCRIBE=true HSA_XNACK=1 \
  LD_PRELOAD=/root/repo/k8s_device_plugin_amd/csrc/libvgpu-hip.so \
  python benchmarks/oversub_bench.py --target-gb 296 --chunk-gb 4 --passes 1 \
  > gpurun_out/oversub_r2c.log 2>&1
echo "oversub rc=$?"
tail -3 gpurun_out/oversub_r2c.log
