/* Fake libhsa-runtime64: hardware-free backend for the HSA-layer hook tests
 * (same pattern as fakehip/).  Implements the memory-pool
 * subset the interceptor hooks, with one GPU-local pool (handle 0xGPU) and
 * one CPU pool (handle 0xCPU).
 *
 * Env:
 *   FAKE_HSA_TOTAL_MEM   bytes in the GPU pool (default 288 GiB)
 */
#define _GNU_SOURCE
#include <pthread.h>
#include <stdint.h>
#include <stdlib.h>
#include <string.h>

typedef int hsa_status_t;
#define HSA_OK 0x0
#define HSA_ERR_OOR 0x1008
#define HSA_ERR_INVALID 0x1001

typedef struct { uint64_t handle; } pool_t;
typedef struct { uint64_t handle; } region_t;

#define GPU_POOL 0x6770ULL /* "gp" */
#define CPU_POOL 0x6370ULL /* "cp" */

static uint64_t g_used;
static pthread_mutex_t g_mu = PTHREAD_MUTEX_INITIALIZER;

static uint64_t total_mem(void) {
  const char *e = getenv("FAKE_HSA_TOTAL_MEM");
  return e ? strtoull(e, NULL, 10) : (288ULL << 30);
}

typedef struct { uint64_t size; uint64_t pool; } hdr_t;

hsa_status_t hsa_init(void) { return HSA_OK; }
hsa_status_t hsa_shut_down(void) { return HSA_OK; }

/* attrs: 0 = SEGMENT (GLOBAL=0), 17 = LOCATION (CPU=0, GPU=1) */
hsa_status_t hsa_amd_memory_pool_get_info(pool_t pool, int attr, void *value) {
  if (!value) return HSA_ERR_INVALID;
  uint32_t *out = (uint32_t *)value;
  if (attr == 0) {
    *out = 0; /* GLOBAL for both pools */
    return HSA_OK;
  }
  if (attr == 17) {
    *out = pool.handle == GPU_POOL ? 1 : 0;
    return HSA_OK;
  }
  return HSA_ERR_INVALID;
}

hsa_status_t hsa_amd_memory_pool_allocate(pool_t pool, size_t size,
                                          uint32_t flags, void **ptr) {
  (void)flags;
  if (!ptr || (pool.handle != GPU_POOL && pool.handle != CPU_POOL))
    return HSA_ERR_INVALID;
  if (pool.handle == GPU_POOL) {
    pthread_mutex_lock(&g_mu);
    if (g_used + size > total_mem()) {
      pthread_mutex_unlock(&g_mu);
      return HSA_ERR_OOR;
    }
    g_used += size;
    pthread_mutex_unlock(&g_mu);
  }
  hdr_t *h = (hdr_t *)malloc(sizeof(hdr_t) + 64);
  if (!h) return HSA_ERR_OOR;
  h->size = size;
  h->pool = pool.handle;
  *ptr = (void *)(h + 1);
  return HSA_OK;
}

hsa_status_t hsa_amd_memory_pool_free(void *ptr) {
  if (!ptr) return HSA_OK;
  hdr_t *h = ((hdr_t *)ptr) - 1;
  if (h->pool == GPU_POOL) {
    pthread_mutex_lock(&g_mu);
    g_used -= h->size;
    pthread_mutex_unlock(&g_mu);
  }
  free(h);
  return HSA_OK;
}

/* legacy region API: attrs 0 = SEGMENT (GLOBAL=0 for both),
 * 1 = GLOBAL_FLAGS (GPU region COARSE_GRAINED=4, CPU FINE_GRAINED=2) —
 * mirrors hsa.h:3219-3262 so the interceptor's classification is testable */
hsa_status_t hsa_region_get_info(region_t region, int attr, void *value) {
  if (!value) return HSA_ERR_INVALID;
  uint32_t *out = (uint32_t *)value;
  if (attr == 0) {
    *out = 0;
    return HSA_OK;
  }
  if (attr == 1) {
    *out = region.handle == GPU_POOL ? 4u : 2u;
    return HSA_OK;
  }
  return HSA_ERR_INVALID;
}

hsa_status_t hsa_memory_allocate(region_t region, size_t size, void **ptr) {
  pool_t p = {region.handle};
  return hsa_amd_memory_pool_allocate(p, size, 0, ptr);
}

hsa_status_t hsa_memory_free(void *ptr) { return hsa_amd_memory_pool_free(ptr); }

/* introspection for tests */
uint64_t fake_hsa_used(void) { return g_used; }
