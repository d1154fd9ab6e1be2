/* HBM memory cap: interposed HIP allocation entry points + usage ledger.
 *
 * Every device allocation is (1) checked against the container's per-device
 * limit summed across ALL processes sharing the region, (2) dispatched to
 * the real allocator — or to hipMallocManaged when the container runs in
 * oversubscription mode (VGPU_OVERSUBSCRIBE=true: the plugin advertises
 * scaled memory and XNACK pages the excess to host DRAM; MI355X equivalent
 * of the reference's CUDA_OVERSUBSCRIBE managed-alloc mode, SURVEY.md §2.6
 * "Oversubscription") — and (3) recorded in a per-process ptr->(size,dev)
 * ledger so hipFree can credit the right device.
 *
 * Reference behavioral spec: cuMemAlloc_v2/cuMemFree_v2/cuMemGetInfo hooks
 * ("device OOM encountered: usage=%lu limit=%lu", SURVEY.md §2.6 "Memory
 * cap"); this is a from-scratch HIP implementation.
 */
#define _GNU_SOURCE
#include "vgpu.h"

#include <fcntl.h>
#include <sys/file.h>
#include <unistd.h>

#include <pthread.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <unistd.h>

typedef int hipError_t;
#define hipSuccess 0
#define hipErrorInvalidValue 1
#define hipErrorOutOfMemory 2

/* ---- allocation ledger: open-addressing hash, process-local ---------- */
#define LEDGER_BUCKETS 65536 /* power of two */
typedef struct {
  void *ptr;
  uint64_t size;
  int32_t dev;
  int32_t live;
} ledger_entry_t;

static ledger_entry_t g_ledger[LEDGER_BUCKETS];
static pthread_mutex_t g_ledger_mu = PTHREAD_MUTEX_INITIALIZER;

static inline size_t lhash(void *p) {
  uintptr_t x = (uintptr_t)p;
  x ^= x >> 17;
  x *= 0xed5ad4bbU;
  x ^= x >> 11;
  return (size_t)(x & (LEDGER_BUCKETS - 1));
}

static int ledger_insert(void *ptr, uint64_t size, int dev) {
  pthread_mutex_lock(&g_ledger_mu);
  size_t i = lhash(ptr);
  for (size_t n = 0; n < LEDGER_BUCKETS; n++, i = (i + 1) & (LEDGER_BUCKETS - 1)) {
    if (!g_ledger[i].live) {
      g_ledger[i] = (ledger_entry_t){ptr, size, dev, 1};
      pthread_mutex_unlock(&g_ledger_mu);
      return 0;
    }
  }
  pthread_mutex_unlock(&g_ledger_mu);
  vgpu_log(VGPU_ERR, "allocation ledger full");
  return -1;
}

static int ledger_remove(void *ptr, uint64_t *size, int *dev) {
  pthread_mutex_lock(&g_ledger_mu);
  size_t i = lhash(ptr);
  for (size_t n = 0; n < LEDGER_BUCKETS; n++, i = (i + 1) & (LEDGER_BUCKETS - 1)) {
    if (g_ledger[i].live && g_ledger[i].ptr == ptr) {
      *size = g_ledger[i].size;
      *dev = g_ledger[i].dev;
      g_ledger[i].live = 0;
      g_ledger[i].ptr = NULL;
      pthread_mutex_unlock(&g_ledger_mu);
      return 0;
    }
    if (!g_ledger[i].live && g_ledger[i].ptr == NULL && g_ledger[i].size == 0 &&
        n > 64)
      break; /* long-gone region; bounded probe */
  }
  pthread_mutex_unlock(&g_ledger_mu);
  return -1;
}

/* ---- cap check ------------------------------------------------------- */
int vgpu_oom_check(int dev, uint64_t request) {
  vgpu_region_t *r = vgpu_region_get();
  if (!r || dev < 0 || dev >= VGPU_MAX_DEVICES) return 0;
  uint64_t limit = r->limit[dev];
  if (limit == 0) return 0;
  vgpu_region_lock(r);
  uint64_t usage = vgpu_region_device_usage(r, dev);
  vgpu_region_unlock(r);
  if (usage + request > limit) {
    vgpu_log(VGPU_ERR,
             "device %d OOM encountered: usage=%llu request=%llu limit=%llu",
             dev, (unsigned long long)usage, (unsigned long long)request,
             (unsigned long long)limit);
    if (getenv(ENV_ACTIVE_OOM_KILLER)) {
      vgpu_log(VGPU_ERR, "ACTIVE_OOM_KILLER set: aborting process");
      _exit(137);
    }
    return -1;
  }
  return 0;
}

uint64_t vgpu_current_usage(int dev) {
  vgpu_region_t *r = vgpu_region_get();
  if (!r) return 0;
  vgpu_region_lock(r);
  uint64_t u = vgpu_region_device_usage(r, dev);
  vgpu_region_unlock(r);
  return u;
}

static void account_alloc(void *ptr, uint64_t size, int dev) {
  vgpu_region_t *r = vgpu_region_get();
  if (r) vgpu_region_add_usage(r, dev, (int64_t)size, 0);
  ledger_insert(ptr, size, dev);
}

static int oversubscribe_mode(void) {
  vgpu_region_t *r = vgpu_region_get();
  return r && r->oversubscribe;
}

/* Host-wide serialization of managed (pageable) allocations across
 * containers (reference unified_lock on /tmp/vgpulock/lock with 1 s wait +
 * expiry strings, SURVEY.md §2.6 "Oversubscription"): concurrent UVM
 * carving from several pods thrashes the migration machinery.  flock is
 * used instead of an expiring lockfile — the kernel releases it when the
 * holder dies, so no stale-lock repair is needed. */
static int unified_lock_acquire(void) {
  int fd = open("/tmp/vgpulock/lock", O_CREAT | O_RDWR, 0666);
  if (fd < 0) return -1; /* lock dir not mounted: proceed unserialized */
  for (int tries = 0; tries < 100; tries++) { /* <= 10 s */
    if (flock(fd, LOCK_EX | LOCK_NB) == 0) return fd;
    usleep(100 * 1000);
  }
  vgpu_log(VGPU_WARN, "unified lock busy >10s; proceeding without it");
  close(fd);
  return -1;
}

static void unified_lock_release(int fd) {
  if (fd >= 0) {
    flock(fd, LOCK_UN);
    close(fd);
  }
}

/* ---- hooks ----------------------------------------------------------- */
typedef hipError_t (*fn_malloc)(void **, size_t);
typedef hipError_t (*fn_malloc_flags)(void **, size_t, unsigned int);
typedef hipError_t (*fn_malloc_async)(void **, size_t, void *);
typedef hipError_t (*fn_free)(void *);
typedef hipError_t (*fn_free_async)(void *, void *);
typedef hipError_t (*fn_malloc_pitch)(void **, size_t *, size_t, size_t);

#define REAL(type, name)                              \
  static type real_##name = NULL;                     \
  if (!real_##name) real_##name = (type)vgpu_real_hip(#name); \
  if (!real_##name) return hipErrorInvalidValue;

static hipError_t alloc_common(void **ptr, size_t size, const char *via,
                               fn_malloc real_fn) {
  vgpu_ensure_initialized();
  int dev = vgpu_current_device();
  if (!vgpu_control_disabled() && vgpu_oom_check(dev, size) != 0)
    return hipErrorOutOfMemory;
  hipError_t e;
  vgpu_tls_passthrough++; /* the HIP runtime allocates via HSA underneath */
  if (!vgpu_control_disabled() && oversubscribe_mode()) {
    /* managed allocation: XNACK pages beyond-HBM working sets to host DRAM;
     * serialized host-wide so co-located pods don't thrash UVM.
     * hipMallocManaged takes THREE arguments — calling it through a 2-arg
     * pointer leaves flags as register garbage (UB). */
    int lk = unified_lock_acquire();
    fn_malloc_flags managed =
        (fn_malloc_flags)vgpu_real_hip("hipMallocManaged");
    e = managed ? managed(ptr, size, 0x01 /* hipMemAttachGlobal */)
                : real_fn(ptr, size);
    unified_lock_release(lk);
  } else {
    e = real_fn(ptr, size);
  }
  vgpu_tls_passthrough--;
  if (e == hipSuccess && !vgpu_control_disabled()) {
    account_alloc(*ptr, size, dev);
    vgpu_log(VGPU_DEBUG, "%s(%zu) dev=%d -> %p", via, size, dev, *ptr);
  }
  return e;
}

hipError_t hipMalloc(void **ptr, size_t size) {
  REAL(fn_malloc, hipMalloc);
  return alloc_common(ptr, size, "hipMalloc", real_hipMalloc);
}

hipError_t hipExtMallocWithFlags(void **ptr, size_t size, unsigned int flags) {
  REAL(fn_malloc_flags, hipExtMallocWithFlags);
  vgpu_ensure_initialized();
  int dev = vgpu_current_device();
  if (!vgpu_control_disabled() && vgpu_oom_check(dev, size) != 0)
    return hipErrorOutOfMemory;
  vgpu_tls_passthrough++;
  hipError_t e = real_hipExtMallocWithFlags(ptr, size, flags);
  vgpu_tls_passthrough--;
  if (e == hipSuccess && !vgpu_control_disabled()) account_alloc(*ptr, size, dev);
  return e;
}

hipError_t hipMallocManaged(void **ptr, size_t size, unsigned int flags) {
  /* managed memory counts against the cap only when NOT oversubscribing:
   * in oversubscribe mode the cap is the scaled (virtual) limit and the
   * check in alloc_common/oom_check still applies through that limit. */
  REAL(fn_malloc_flags, hipMallocManaged);
  vgpu_ensure_initialized();
  int dev = vgpu_current_device();
  if (!vgpu_control_disabled() && vgpu_oom_check(dev, size) != 0)
    return hipErrorOutOfMemory;
  vgpu_tls_passthrough++;
  int lk = (!vgpu_control_disabled() && oversubscribe_mode())
               ? unified_lock_acquire() : -1;
  hipError_t e = real_hipMallocManaged(ptr, size, flags);
  unified_lock_release(lk);
  vgpu_tls_passthrough--;
  if (e == hipSuccess && !vgpu_control_disabled()) account_alloc(*ptr, size, dev);
  return e;
}

hipError_t hipMallocAsync(void **ptr, size_t size, void *stream) {
  REAL(fn_malloc_async, hipMallocAsync);
  vgpu_ensure_initialized();
  int dev = vgpu_current_device();
  if (!vgpu_control_disabled() && vgpu_oom_check(dev, size) != 0)
    return hipErrorOutOfMemory;
  vgpu_tls_passthrough++;
  hipError_t e = real_hipMallocAsync(ptr, size, stream);
  vgpu_tls_passthrough--;
  if (e == hipSuccess && !vgpu_control_disabled()) account_alloc(*ptr, size, dev);
  return e;
}

hipError_t hipMallocFromPoolAsync(void **ptr, size_t size, void *pool,
                                  void *stream) {
  typedef hipError_t (*fn)(void **, size_t, void *, void *);
  REAL(fn, hipMallocFromPoolAsync);
  vgpu_ensure_initialized();
  int dev = vgpu_current_device();
  if (!vgpu_control_disabled() && vgpu_oom_check(dev, size) != 0)
    return hipErrorOutOfMemory;
  vgpu_tls_passthrough++;
  hipError_t e = real_hipMallocFromPoolAsync(ptr, size, pool, stream);
  vgpu_tls_passthrough--;
  if (e == hipSuccess && !vgpu_control_disabled()) account_alloc(*ptr, size, dev);
  return e;
}

hipError_t hipMallocPitch(void **ptr, size_t *pitch, size_t width,
                          size_t height) {
  REAL(fn_malloc_pitch, hipMallocPitch);
  vgpu_ensure_initialized();
  int dev = vgpu_current_device();
  /* conservative pre-check with 256-aligned pitch; re-account with the
   * real pitch after the call */
  uint64_t est = ((width + 255) & ~255ULL) * height;
  if (!vgpu_control_disabled() && vgpu_oom_check(dev, est) != 0)
    return hipErrorOutOfMemory;
  vgpu_tls_passthrough++;
  hipError_t e = real_hipMallocPitch(ptr, pitch, width, height);
  vgpu_tls_passthrough--;
  if (e == hipSuccess && !vgpu_control_disabled())
    account_alloc(*ptr, (uint64_t)(*pitch) * height, dev);
  return e;
}

static void account_free(void *ptr) {
  if (!ptr || vgpu_control_disabled()) return;
  uint64_t size;
  int dev;
  if (ledger_remove(ptr, &size, &dev) == 0) {
    vgpu_region_t *r = vgpu_region_get();
    if (r) vgpu_region_add_usage(r, dev, -(int64_t)size, 0);
  }
}

hipError_t hipFree(void *ptr) {
  REAL(fn_free, hipFree);
  vgpu_ensure_initialized();
  vgpu_tls_passthrough++;
  hipError_t e = real_hipFree(ptr);
  vgpu_tls_passthrough--;
  if (e == hipSuccess) account_free(ptr);
  return e;
}

hipError_t hipFreeAsync(void *ptr, void *stream) {
  REAL(fn_free_async, hipFreeAsync);
  vgpu_ensure_initialized();
  vgpu_tls_passthrough++;
  hipError_t e = real_hipFreeAsync(ptr, stream);
  vgpu_tls_passthrough--;
  if (e == hipSuccess) account_free(ptr);
  return e;
}

/* ---- array allocations (the reference accounts cuArray{,3D}Create /
 * cuMipmappedArrayCreate the same way, SURVEY.md §2.6 "Memory cap") ---- */
typedef struct { size_t width, height, depth; } vgpu_extent_t;
typedef struct { void *ptr; size_t pitch, xsize, ysize; } vgpu_pitched_ptr_t;
typedef struct { int x, y, z, w; int f; } vgpu_chan_desc_t;

static uint64_t chan_bytes(const vgpu_chan_desc_t *d) {
  if (!d) return 4;
  int bits = d->x + d->y + d->z + d->w;
  return bits > 0 ? (uint64_t)(bits + 7) / 8 : 4;
}

hipError_t hipMalloc3D(vgpu_pitched_ptr_t *p, vgpu_extent_t extent) {
  typedef hipError_t (*fn)(vgpu_pitched_ptr_t *, vgpu_extent_t);
  REAL(fn, hipMalloc3D);
  vgpu_ensure_initialized();
  int dev = vgpu_current_device();
  uint64_t est = ((extent.width + 255) & ~255ULL) * extent.height *
                 (extent.depth ? extent.depth : 1);
  if (!vgpu_control_disabled() && vgpu_oom_check(dev, est) != 0)
    return hipErrorOutOfMemory;
  vgpu_tls_passthrough++;
  hipError_t e = real_hipMalloc3D(p, extent);
  vgpu_tls_passthrough--;
  if (e == hipSuccess && !vgpu_control_disabled())
    account_alloc(p->ptr,
                  p->pitch * extent.height * (extent.depth ? extent.depth : 1),
                  dev);
  return e;
}

hipError_t hipMallocArray(void **array, const vgpu_chan_desc_t *desc,
                          size_t width, size_t height, unsigned int flags) {
  typedef hipError_t (*fn)(void **, const vgpu_chan_desc_t *, size_t, size_t,
                           unsigned int);
  REAL(fn, hipMallocArray);
  vgpu_ensure_initialized();
  int dev = vgpu_current_device();
  uint64_t bytes = chan_bytes(desc) * width * (height ? height : 1);
  if (!vgpu_control_disabled() && vgpu_oom_check(dev, bytes) != 0)
    return hipErrorOutOfMemory;
  vgpu_tls_passthrough++;
  hipError_t e = real_hipMallocArray(array, desc, width, height, flags);
  vgpu_tls_passthrough--;
  if (e == hipSuccess && !vgpu_control_disabled())
    account_alloc(*array, bytes, dev);
  return e;
}

hipError_t hipMalloc3DArray(void **array, const vgpu_chan_desc_t *desc,
                            vgpu_extent_t extent, unsigned int flags) {
  typedef hipError_t (*fn)(void **, const vgpu_chan_desc_t *, vgpu_extent_t,
                           unsigned int);
  REAL(fn, hipMalloc3DArray);
  vgpu_ensure_initialized();
  int dev = vgpu_current_device();
  uint64_t bytes = chan_bytes(desc) * extent.width *
                   (extent.height ? extent.height : 1) *
                   (extent.depth ? extent.depth : 1);
  if (!vgpu_control_disabled() && vgpu_oom_check(dev, bytes) != 0)
    return hipErrorOutOfMemory;
  vgpu_tls_passthrough++;
  hipError_t e = real_hipMalloc3DArray(array, desc, extent, flags);
  vgpu_tls_passthrough--;
  if (e == hipSuccess && !vgpu_control_disabled())
    account_alloc(*array, bytes, dev);
  return e;
}

hipError_t hipFreeArray(void *array) {
  typedef hipError_t (*fn)(void *);
  REAL(fn, hipFreeArray);
  vgpu_ensure_initialized();
  vgpu_tls_passthrough++;
  hipError_t e = real_hipFreeArray(array);
  vgpu_tls_passthrough--;
  if (e == hipSuccess) account_free(array);
  return e;
}
