/* PLT-linked HSA consumer: exercises the HSA-layer quota path the way an
 * HSA-direct library would (no HIP).  Linked against the fake
 * libhsa-runtime64; run under LD_PRELOAD libvgpu-hip.so with
 * VGPU_REAL_HSA_PATH pointing at the fake.
 *
 * argv: alloc_mb [alloc_mb...]   allocate each in order from the GPU pool;
 * prints "ok <ptr-index>" or "oom <ptr-index>" per allocation, then usage.
 * First arg "legacy" switches to the legacy hsa_memory_allocate API on the
 * GPU region; "legacycpu" uses the host (fine-grained) region, which must
 * stay uncounted.
 */
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

typedef int hsa_status_t;
typedef struct { uint64_t handle; } pool_t;
typedef struct { uint64_t handle; } region_t;

extern hsa_status_t hsa_amd_memory_pool_allocate(pool_t, size_t, uint32_t,
                                                 void **);
extern hsa_status_t hsa_amd_memory_pool_free(void *);
extern hsa_status_t hsa_memory_allocate(region_t, size_t, void **);

int main(int argc, char **argv) {
  pool_t gpu = {0x6770};
  region_t gpu_region = {0x6770}, cpu_region = {0x6370};
  void *ptrs[64] = {0};
  int n = 0;
  int start = 1, legacy = 0, legacy_cpu = 0;
  if (argc > 1 && strcmp(argv[1], "legacy") == 0) { legacy = 1; start = 2; }
  if (argc > 1 && strcmp(argv[1], "legacycpu") == 0) {
    legacy = legacy_cpu = 1;
    start = 2;
  }
  for (int i = start; i < argc && n < 64; i++, n++) {
    size_t mb = strtoull(argv[i], NULL, 10);
    hsa_status_t s;
    if (legacy)
      s = hsa_memory_allocate(legacy_cpu ? cpu_region : gpu_region, mb << 20,
                              &ptrs[n]);
    else
      s = hsa_amd_memory_pool_allocate(gpu, mb << 20, 0, &ptrs[n]);
    printf("%s %d\n", s == 0 ? "ok" : "oom", n);
  }
  /* free the even ones to test ledger removal */
  for (int i = 0; i < n; i += 2)
    if (ptrs[i]) hsa_amd_memory_pool_free(ptrs[i]);
  printf("done\n");
  fflush(stdout);
  return 0;
}
