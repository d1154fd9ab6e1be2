/* PLT-linked HSA consumer: exercises the HSA-layer quota path the way an
 * HSA-direct library would (no HIP).  Linked against the fake
 * libhsa-runtime64; run under LD_PRELOAD libvgpu-hip.so with
 * VGPU_REAL_HSA_PATH pointing at the fake.
 *
 * argv: alloc_mb [alloc_mb...]   allocate each in order from the GPU pool;
 * prints "ok <ptr-index>" or "oom <ptr-index>" per allocation, then usage.
 */
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>

typedef int hsa_status_t;
typedef struct { uint64_t handle; } pool_t;

extern hsa_status_t hsa_amd_memory_pool_allocate(pool_t, size_t, uint32_t,
                                                 void **);
extern hsa_status_t hsa_amd_memory_pool_free(void *);

int main(int argc, char **argv) {
  pool_t gpu = {0x6770};
  void *ptrs[64] = {0};
  int n = 0;
  for (int i = 1; i < argc && n < 64; i++, n++) {
    size_t mb = strtoull(argv[i], NULL, 10);
    hsa_status_t s = hsa_amd_memory_pool_allocate(gpu, mb << 20, 0, &ptrs[n]);
    printf("%s %d\n", s == 0 ? "ok" : "oom", n);
  }
  /* free the even ones to test ledger removal */
  for (int i = 0; i < n; i += 2)
    if (ptrs[i]) hsa_amd_memory_pool_free(ptrs[i]);
  printf("done\n");
  fflush(stdout);
  return 0;
}
