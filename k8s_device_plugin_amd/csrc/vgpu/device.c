/* Device-info virtualization: the container sees its quota, not the card.
 *
 * Reference behavioral spec: cuMemGetInfo_v2 reports (limit-usage, limit)
 * and cuDeviceTotalMem the quota ("lies to frameworks");
 * tools get the same view via the SMI spoof (smi.c).  MI355X: a whole card
 * reports 288 GB HBM3E; a 4-way split reports 72 GB.
 *
 * HIP versioned-symbol note: since ROCm 6 the public hipGetDeviceProperties
 * is a macro for hipGetDevicePropertiesR0600 (hip_runtime_api.h:103); we
 * export BOTH the R0600 symbol (current ABI) and the legacy unversioned one
 * (pre-6.0 binaries), patching totalGlobalMem at the right offset for each
 * struct generation.
 */
#define _GNU_SOURCE
#include "vgpu.h"

#include <dlfcn.h>
#include <stddef.h>
#include <string.h>

typedef int hipError_t;
#define hipSuccess 0
#define hipErrorInvalidValue 1

/* layout prefix of hipDeviceProp_tR0600 (hip_runtime_api.h:110-116) */
typedef struct {
  char name[256];
  char uuid[16];
  char luid[8];
  unsigned int luidDeviceNodeMask;
  size_t totalGlobalMem;
} prop_r0600_prefix_t;

/* layout prefix of the legacy (R0000) hipDeviceProp_t */
typedef struct {
  char name[256];
  size_t totalGlobalMem;
} prop_r0000_prefix_t;

/* The current device is consulted on EVERY launch (limiter gate) and every
 * allocation; calling the real hipGetDevice each time costs a runtime API
 * round-trip per kernel.  hipSetDevice is interposed below to keep a TLS
 * shadow, so the hot path is one TLS read. */
static __thread int tls_current_dev = -1;

hipError_t hipSetDevice(int device) {
  typedef hipError_t (*fn)(int);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hip("hipSetDevice");
  if (!real) return hipErrorInvalidValue;
  hipError_t e = real(device);
  if (e == hipSuccess) tls_current_dev = device;
  return e;
}

int vgpu_current_device(void) {
  if (tls_current_dev >= 0) return tls_current_dev;
  typedef hipError_t (*fn)(int *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hip("hipGetDevice");
  int dev = 0;
  if (real) real(&dev);
  tls_current_dev = dev;
  return dev;
}

hipError_t hipMemGetInfo(size_t *free_out, size_t *total_out) {
  typedef hipError_t (*fn)(size_t *, size_t *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hip("hipMemGetInfo");
  if (!real) return hipErrorInvalidValue;
  vgpu_ensure_initialized();
  size_t real_free = 0, real_total = 0;
  hipError_t e = real(&real_free, &real_total);
  if (e != hipSuccess) return e;
  int dev = vgpu_current_device();
  uint64_t limit = vgpu_control_disabled() ? 0 : vgpu_region_limit(dev);
  if (limit == 0) {
    if (free_out) *free_out = real_free;
    if (total_out) *total_out = real_total;
    return e;
  }
  uint64_t usage = vgpu_current_usage(dev);
  uint64_t lim_free = usage >= limit ? 0 : limit - usage;
  /* Co-located pods can leave the card with less physical free than this
   * container's remaining quota; promising the larger number makes
   * frameworks over-allocate and hit hard OOM.  Clamp to the real free —
   * except in oversubscribe mode, where exceeding physical HBM is the
   * whole point (managed memory pages to host DRAM). */
  vgpu_region_t *vr = vgpu_region_get();
  int oversub = vr && vr->oversubscribe;
  if (!oversub && lim_free > real_free) lim_free = real_free;
  vgpu_log(VGPU_DEBUG,
           "hipMemGetInfo: orig free=%zu total=%zu limit=%llu usage=%llu",
           real_free, real_total, (unsigned long long)limit,
           (unsigned long long)usage);
  if (free_out) *free_out = (size_t)lim_free;
  if (total_out) *total_out = (size_t)limit;
  return e;
}

hipError_t hipDeviceTotalMem(size_t *bytes, int device) {
  typedef hipError_t (*fn)(size_t *, int);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_hip("hipDeviceTotalMem");
  if (!real) return hipErrorInvalidValue;
  vgpu_ensure_initialized();
  hipError_t e = real(bytes, device);
  if (e != hipSuccess) return e;
  uint64_t limit = vgpu_control_disabled() ? 0 : vgpu_region_limit(device);
  if (limit && bytes) *bytes = (size_t)limit;
  return e;
}

static hipError_t get_props(void *prop, int device, const char *sym,
                            size_t mem_offset) {
  typedef hipError_t (*fn)(void *, int);
  fn real = (fn)vgpu_real_hip(sym);
  if (!real) return hipErrorInvalidValue;
  vgpu_ensure_initialized();
  hipError_t e = real(prop, device);
  if (e != hipSuccess) return e;
  uint64_t limit = vgpu_control_disabled() ? 0 : vgpu_region_limit(device);
  if (limit && prop)
    memcpy((char *)prop + mem_offset, &limit, sizeof(uint64_t));
  return e;
}

hipError_t hipGetDevicePropertiesR0600(void *prop, int device) {
  return get_props(prop, device, "hipGetDevicePropertiesR0600",
                   offsetof(prop_r0600_prefix_t, totalGlobalMem));
}

hipError_t hipGetDeviceProperties(void *prop, int device) {
  return get_props(prop, device, "hipGetDeviceProperties",
                   offsetof(prop_r0000_prefix_t, totalGlobalMem));
}

/* CUDA-11.3-style dynamic resolution (the reference hooks cuGetProcAddress
 * for this, SURVEY.md §2.6 "dlsym bootstrap"): resolve through the global
 * scope, where this preloaded library shadows libamdhip64. */
hipError_t hipGetProcAddress(const char *symbol, void **pfn, int hipVersion,
                             uint64_t flags, void *symbolStatus) {
  if (!symbol || !pfn) return hipErrorInvalidValue;
  void *p = dlsym(RTLD_DEFAULT, symbol);
  if (!p) p = vgpu_real_hip(symbol);
  if (!p) {
    typedef hipError_t (*fn)(const char *, void **, int, uint64_t, void *);
    fn real = (fn)vgpu_real_hip("hipGetProcAddress");
    if (real) return real(symbol, pfn, hipVersion, flags, symbolStatus);
    return hipErrorInvalidValue;
  }
  *pfn = p;
  return hipSuccess;
}
