#!/bin/bash
# Round-2 GPU call 3: partition/XNACK probes, density fairness rerun
# (per-pod MIOpen copies), bounded beyond-physical paging test, oversub
# bench with correct target, rocprof kernel-stats evidence.
set -x
cd /root/repo

# 1. Why did the CPX switch fail? Probe the sysfs surface directly.
{
  echo "=== rocminfo xnack ==="
  /opt/rocm/bin/rocminfo 2>/dev/null | grep -i -m4 "xnack\|gfx"
  echo "=== partition files ==="
  ls -la /sys/class/drm/card*/device/current_compute_partition 2>&1
  for f in /sys/class/drm/card*/device/current_compute_partition; do
    [ -e "$f" ] || continue
    echo "--- $f"
    cat "$f"
    echo "available:"; cat "$(dirname "$f")/available_compute_partition" 2>&1
    orig=$(cat "$f")
    echo "try CPX:"
    if echo CPX > "$f" 2>&1; then
      echo "write ok, now: $(cat "$f")"
      echo "KFD gpu nodes now:"
      grep -l . /sys/class/kfd/kfd/topology/nodes/*/gpu_id 2>/dev/null | wc -l
      ls /dev/dri/ | tr '\n' ' '; echo
      echo "restore $orig:"
      echo "$orig" > "$f" 2>&1 && echo "restored: $(cat "$f")"
    else
      echo "write failed: $?"
      dmesg 2>/dev/null | tail -5
    fi
  done
} > gpurun_out/partition_probe_r2.log 2>&1
tail -30 gpurun_out/partition_probe_r2.log

# 2. Density fairness rerun with per-pod MIOpen db copies
timeout 500 python bench.py --density-only --density-seconds 20 \
  > gpurun_out/density_r2b.log 2>&1
echo "density rc=$?"
tail -2 gpurun_out/density_r2b.log

# 3. Revised beyond-physical paging test
timeout 700 python -m pytest "tests/test_gpu.py::TestOversubscriptionReal::test_beyond_physical_hbm_pages_and_computes" -q -rs \
  > gpurun_out/paging_test_r2.log 2>&1
echo "paging rc=$?"
tail -6 gpurun_out/paging_test_r2.log

# 4. Oversubscription bench, correctly sized (310 GB on the 288 GB card)
timeout 500 env \
  VGPU_DEVICE_MEMORY_SHARED_CACHE=/tmp/oversub.cache \
  VGPU_DEVICE_MEMORY_LIMIT=409600m VGPU_OVERSUBSCRIBE=true HSA_XNACK=1 \
  LD_PRELOAD=/root/repo/k8s_device_plugin_amd/csrc/libvgpu-hip.so \
  python benchmarks/oversub_bench.py --target-gb 310 --chunk-gb 2 --passes 1 \
  > gpurun_out/oversub_r2b.log 2>&1
echo "oversub rc=$?"
tail -2 gpurun_out/oversub_r2b.log

# 5. rocprof kernel stats of a short preloaded run (solver evidence:
#    tuned igemm/winograd kernels, no naive_conv fallback)
cd /tmp && export TMPDIR=/tmp
cp -r /root/repo/miopen_udb /tmp/mdb
cp -r /root/repo/miopen_cache /tmp/mcache
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_r2 -o bench_vgpu -- \
  env VGPU_DEVICE_MEMORY_SHARED_CACHE=/tmp/prof.cache \
  VGPU_DEVICE_MEMORY_LIMIT=147448m \
  LD_PRELOAD=/root/repo/k8s_device_plugin_amd/csrc/libvgpu-hip.so \
  MIOPEN_USER_DB_PATH=/tmp/mdb MIOPEN_CUSTOM_CACHE_DIR=/tmp/mcache \
  python /root/repo/bench.py --worker --cases resnet50_inf,resnet50_train,vgg16_inf,deeplab_train --steps 3 --warmup 1 < <(echo GO) \
  > /root/repo/gpurun_out/prof_r2_run.log 2>&1
echo "rocprof rc=$?"
ls /root/repo/gpurun_out/prof_r2 2>/dev/null | head
