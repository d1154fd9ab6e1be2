/* HSA-layer memory hook.
 *
 * PyTorch and most apps allocate through HIP, but some libraries (and parts
 * of the ROCm stack) call the HSA runtime directly —
 * hsa_amd_memory_pool_allocate / hsa_memory_allocate — which would bypass
 * the HIP-layer ledger (SURVEY.md §7 hard part 1; the reference's analogue
 * is its cuGetProcAddress problem).  This file interposes the HSA entry
 * points and keeps the SAME shared-region ledger consistent:
 *
 *  - allocations made *by* our HIP wrappers (libamdhip64 internally calls
 *    hsa_amd_memory_pool_allocate) are NOT double-counted: the HIP wrapper
 *    sets vgpu_tls_passthrough around the real call;
 *  - only device-local pools count (segment GLOBAL + location GPU, queried
 *    once per pool via the real hsa_amd_memory_pool_get_info and cached);
 *  - frees look the pointer up in a local ledger, so host-pool frees and
 *    unknown pointers pass through untouched.
 *
 * Types/enum values below mirror the stable public HSA ABI
 * (/opt/rocm/include/hsa/hsa.h:126-170, hsa_ext_amd.h:1403-1551); they are
 * re-declared locally so the CPU CI build needs no ROCm headers.
 */
#define _GNU_SOURCE
#include "vgpu.h"

#include <pthread.h>
#include <stdint.h>
#include <string.h>

typedef int hsa_status_t;
#define HSA_STATUS_SUCCESS 0x0
#define HSA_STATUS_ERROR_OUT_OF_RESOURCES 0x1008

typedef struct { uint64_t handle; } hsa_amd_memory_pool_t;
typedef struct { uint64_t handle; } hsa_region_t;

/* hsa_ext_amd.h attribute/enum values (verified against ROCm 7.2) */
#define POOL_INFO_SEGMENT 0            /* HSA_AMD_MEMORY_POOL_INFO_SEGMENT */
#define POOL_INFO_LOCATION 17          /* HSA_AMD_MEMORY_POOL_INFO_LOCATION */
#define SEGMENT_GLOBAL 0               /* HSA_AMD_SEGMENT_GLOBAL */
#define LOCATION_GPU 1                 /* HSA_AMD_MEMORY_POOL_LOCATION_GPU */

/* ---- per-pool device-local classification cache ---------------------- */
#define POOL_CACHE 64
static struct { uint64_t handle; int device_local; } g_pools[POOL_CACHE];
static int g_pool_count = 0;
static pthread_mutex_t g_pool_mu = PTHREAD_MUTEX_INITIALIZER;

static int pool_is_device_local(hsa_amd_memory_pool_t pool) {
  pthread_mutex_lock(&g_pool_mu);
  for (int i = 0; i < g_pool_count; i++) {
    if (g_pools[i].handle == pool.handle) {
      int r = g_pools[i].device_local;
      pthread_mutex_unlock(&g_pool_mu);
      return r;
    }
  }
  pthread_mutex_unlock(&g_pool_mu);

  typedef hsa_status_t (*fn)(hsa_amd_memory_pool_t, int, void *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hsa("hsa_amd_memory_pool_get_info");
  int local = 0;
  if (real) {
    uint32_t segment = ~0u, location = ~0u;
    vgpu_tls_passthrough++;
    hsa_status_t s1 = real(pool, POOL_INFO_SEGMENT, &segment);
    hsa_status_t s2 = real(pool, POOL_INFO_LOCATION, &location);
    vgpu_tls_passthrough--;
    if (s1 == HSA_STATUS_SUCCESS && segment == SEGMENT_GLOBAL &&
        (s2 != HSA_STATUS_SUCCESS || location == LOCATION_GPU))
      local = 1;
  }
  pthread_mutex_lock(&g_pool_mu);
  if (g_pool_count < POOL_CACHE) {
    g_pools[g_pool_count].handle = pool.handle;
    g_pools[g_pool_count].device_local = local;
    g_pool_count++;
  }
  pthread_mutex_unlock(&g_pool_mu);
  return local;
}

/* ---- hsa-side pointer ledger ----------------------------------------- */
#define HSA_LEDGER 4096
static struct { void *ptr; uint64_t size; int dev; } g_hsa_ledger[HSA_LEDGER];
static pthread_mutex_t g_hsa_mu = PTHREAD_MUTEX_INITIALIZER;

static void hsa_ledger_insert(void *ptr, uint64_t size, int dev) {
  pthread_mutex_lock(&g_hsa_mu);
  for (int i = 0; i < HSA_LEDGER; i++) {
    if (g_hsa_ledger[i].ptr == NULL) {
      g_hsa_ledger[i].ptr = ptr;
      g_hsa_ledger[i].size = size;
      g_hsa_ledger[i].dev = dev;
      break;
    }
  }
  pthread_mutex_unlock(&g_hsa_mu);
}

static int hsa_ledger_remove(void *ptr, uint64_t *size, int *dev) {
  int found = 0;
  pthread_mutex_lock(&g_hsa_mu);
  for (int i = 0; i < HSA_LEDGER; i++) {
    if (g_hsa_ledger[i].ptr == ptr) {
      *size = g_hsa_ledger[i].size;
      *dev = g_hsa_ledger[i].dev;
      g_hsa_ledger[i].ptr = NULL;
      found = 1;
      break;
    }
  }
  pthread_mutex_unlock(&g_hsa_mu);
  return found;
}

/* ---- hooks ------------------------------------------------------------ */
hsa_status_t hsa_amd_memory_pool_allocate(hsa_amd_memory_pool_t pool,
                                          size_t size, uint32_t flags,
                                          void **ptr) {
  typedef hsa_status_t (*fn)(hsa_amd_memory_pool_t, size_t, uint32_t, void **);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hsa("hsa_amd_memory_pool_allocate");
  if (!real) return HSA_STATUS_ERROR_OUT_OF_RESOURCES;
  if (vgpu_control_disabled() || vgpu_tls_passthrough)
    return real(pool, size, flags, ptr);

  vgpu_ensure_initialized();
  int counted = pool_is_device_local(pool);
  int dev = counted ? vgpu_current_device() : -1;
  if (counted && vgpu_oom_check(dev, size) != 0) {
    vgpu_log(VGPU_WARN, "hsa pool alloc %zu over limit dev=%d", size, dev);
    return HSA_STATUS_ERROR_OUT_OF_RESOURCES;
  }
  vgpu_tls_passthrough++;
  hsa_status_t s = real(pool, size, flags, ptr);
  vgpu_tls_passthrough--;
  if (s == HSA_STATUS_SUCCESS && counted) {
    vgpu_region_t *r = vgpu_region_get();
    if (r) vgpu_region_add_usage(r, dev, (int64_t)size, 0);
    hsa_ledger_insert(*ptr, size, dev);
    vgpu_log(VGPU_DEBUG, "hsa_amd_memory_pool_allocate(%zu) dev=%d -> %p",
             size, dev, *ptr);
  }
  return s;
}

hsa_status_t hsa_amd_memory_pool_free(void *ptr) {
  typedef hsa_status_t (*fn)(void *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hsa("hsa_amd_memory_pool_free");
  if (!real) return HSA_STATUS_ERROR_OUT_OF_RESOURCES;
  if (vgpu_control_disabled() || vgpu_tls_passthrough) return real(ptr);

  uint64_t size;
  int dev;
  vgpu_tls_passthrough++;
  hsa_status_t s = real(ptr);
  vgpu_tls_passthrough--;
  if (s == HSA_STATUS_SUCCESS && ptr && hsa_ledger_remove(ptr, &size, &dev)) {
    vgpu_region_t *r = vgpu_region_get();
    if (r) vgpu_region_add_usage(r, dev, -(int64_t)size, 0);
  }
  return s;
}

/* Legacy region-based API: device HBM shows up here as a GLOBAL-segment
 * COARSE_GRAINED region (hsa.h:3219-3262 — host staging regions are
 * FINE_GRAINED/KERNARG), so classify by hsa_region_get_info and charge
 * coarse-grained global regions to the ledger like the pool path.  The
 * dominant HSA consumers (HIP runtime, RCCL staging) use the pool API;
 * this closes the quota bypass for direct legacy-API callers. */
#define REGION_INFO_SEGMENT 0      /* HSA_REGION_INFO_SEGMENT */
#define REGION_INFO_GLOBAL_FLAGS 1 /* HSA_REGION_INFO_GLOBAL_FLAGS */
#define REGION_SEGMENT_GLOBAL 0    /* HSA_REGION_SEGMENT_GLOBAL */
#define REGION_FLAG_COARSE 4       /* HSA_REGION_GLOBAL_FLAG_COARSE_GRAINED */

static struct { uint64_t handle; int device_local; } g_regions[POOL_CACHE];
static int g_region_count = 0;

static int legacy_region_is_device_local(hsa_region_t region) {
  pthread_mutex_lock(&g_pool_mu);
  for (int i = 0; i < g_region_count; i++) {
    if (g_regions[i].handle == region.handle) {
      int r = g_regions[i].device_local;
      pthread_mutex_unlock(&g_pool_mu);
      return r;
    }
  }
  pthread_mutex_unlock(&g_pool_mu);

  typedef hsa_status_t (*fn)(hsa_region_t, int, void *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hsa("hsa_region_get_info");
  int local = 0;
  if (real) {
    uint32_t segment = ~0u, flags = 0;
    vgpu_tls_passthrough++;
    hsa_status_t s1 = real(region, REGION_INFO_SEGMENT, &segment);
    hsa_status_t s2 = real(region, REGION_INFO_GLOBAL_FLAGS, &flags);
    vgpu_tls_passthrough--;
    if (s1 == HSA_STATUS_SUCCESS && segment == REGION_SEGMENT_GLOBAL &&
        s2 == HSA_STATUS_SUCCESS && (flags & REGION_FLAG_COARSE))
      local = 1;
  }
  pthread_mutex_lock(&g_pool_mu);
  if (g_region_count < POOL_CACHE) {
    g_regions[g_region_count].handle = region.handle;
    g_regions[g_region_count].device_local = local;
    g_region_count++;
  }
  pthread_mutex_unlock(&g_pool_mu);
  return local;
}

hsa_status_t hsa_memory_allocate(hsa_region_t region, size_t size, void **ptr) {
  typedef hsa_status_t (*fn)(hsa_region_t, size_t, void **);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hsa("hsa_memory_allocate");
  if (!real) return HSA_STATUS_ERROR_OUT_OF_RESOURCES;
  if (vgpu_control_disabled() || vgpu_tls_passthrough)
    return real(region, size, ptr);
  vgpu_ensure_initialized();
  int counted = legacy_region_is_device_local(region);
  int dev = counted ? vgpu_current_device() : -1;
  if (counted && vgpu_oom_check(dev, size) != 0) {
    vgpu_log(VGPU_WARN, "hsa region alloc %zu over limit dev=%d", size, dev);
    return HSA_STATUS_ERROR_OUT_OF_RESOURCES;
  }
  vgpu_tls_passthrough++;
  hsa_status_t s = real(region, size, ptr);
  vgpu_tls_passthrough--;
  if (s == HSA_STATUS_SUCCESS && counted) {
    vgpu_region_t *r = vgpu_region_get();
    if (r) vgpu_region_add_usage(r, dev, (int64_t)size, 0);
    hsa_ledger_insert(*ptr, size, dev);
    vgpu_log(VGPU_DEBUG, "hsa_memory_allocate(%zu) dev=%d -> %p", size, dev,
             *ptr);
  }
  return s;
}

hsa_status_t hsa_memory_free(void *ptr) {
  typedef hsa_status_t (*fn)(void *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hsa("hsa_memory_free");
  if (!real) return HSA_STATUS_ERROR_OUT_OF_RESOURCES;
  if (vgpu_control_disabled() || vgpu_tls_passthrough) return real(ptr);
  uint64_t size;
  int dev;
  vgpu_tls_passthrough++;
  hsa_status_t s = real(ptr);
  vgpu_tls_passthrough--;
  if (s == HSA_STATUS_SUCCESS && ptr && hsa_ledger_remove(ptr, &size, &dev)) {
    vgpu_region_t *r = vgpu_region_get();
    if (r) vgpu_region_add_usage(r, dev, -(int64_t)size, 0);
  }
  return s;
}
