/* amd-smi spoofing: interpose libamd_smi so `amd-smi` inside a container
 * reports the vGPU quota, not the physical card — the modern-tool
 * counterpart of smi.c's rocm-smi (librocm_smi64) spoofing; together they
 * are the MI355X analog of the reference's ~260 nvml* exports feeding
 * nvidia-smi (SURVEY.md §2.6 "Tool spoofing").
 *
 * amdsmi is handle-based: amdsmi_get_processor_handles() is interposed to
 * record each handle's global enumeration index, which then maps to a
 * visible-device slot exactly like an rsmi device index
 * (vgpu_smi_index_to_vdev, smi.c).  Types mirror the stable public ABI
 * (/opt/rocm/include/amd_smi/amdsmi.h: amdsmi_vram_usage_t 684-688,
 * status 304, VRAM mem type 1570).
 */
#define _GNU_SOURCE
#include "vgpu.h"

#include <pthread.h>
#include <stdint.h>
#include <string.h>

typedef int amdsmi_status_t;
#define AMDSMI_OK 0
typedef void *amdsmi_processor_handle;
typedef void *amdsmi_socket_handle;
typedef int amdsmi_memory_type_t; /* AMDSMI_MEM_TYPE_VRAM = 0 */

typedef struct {
  uint32_t vram_total; /* MB */
  uint32_t vram_used;  /* MB */
  uint32_t reserved[2];
} amdsmi_vram_usage_t;

/* ---- handle -> global index registry -------------------------------- */
#define MAX_HANDLES 64
static amdsmi_processor_handle g_handles[MAX_HANDLES];
static int g_handle_count = 0;
static pthread_mutex_t g_h_mu = PTHREAD_MUTEX_INITIALIZER;

static void record_handles(amdsmi_processor_handle *hs, uint32_t n) {
  pthread_mutex_lock(&g_h_mu);
  for (uint32_t i = 0; i < n; i++) {
    int known = 0;
    for (int j = 0; j < g_handle_count; j++)
      if (g_handles[j] == hs[i]) { known = 1; break; }
    if (!known && g_handle_count < MAX_HANDLES)
      g_handles[g_handle_count++] = hs[i];
  }
  pthread_mutex_unlock(&g_h_mu);
}

static int handle_to_vdev(amdsmi_processor_handle h) {
  int idx = -1;
  pthread_mutex_lock(&g_h_mu);
  for (int j = 0; j < g_handle_count; j++)
    if (g_handles[j] == h) { idx = j; break; }
  pthread_mutex_unlock(&g_h_mu);
  if (idx < 0) return -1;
  return vgpu_smi_index_to_vdev((uint32_t)idx);
}

/* ---- hooks ------------------------------------------------------------ */
amdsmi_status_t amdsmi_get_processor_handles(amdsmi_socket_handle socket,
                                             uint32_t *count,
                                             amdsmi_processor_handle *handles) {
  typedef amdsmi_status_t (*fn)(amdsmi_socket_handle, uint32_t *,
                                amdsmi_processor_handle *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_amdsmi("amdsmi_get_processor_handles");
  if (!real) return 1;
  amdsmi_status_t s = real(socket, count, handles);
  if (s == AMDSMI_OK && handles && count) record_handles(handles, *count);
  return s;
}

amdsmi_status_t amdsmi_get_gpu_memory_total(amdsmi_processor_handle h,
                                            amdsmi_memory_type_t type,
                                            uint64_t *total) {
  typedef amdsmi_status_t (*fn)(amdsmi_processor_handle, amdsmi_memory_type_t,
                                uint64_t *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_amdsmi("amdsmi_get_gpu_memory_total");
  if (!real) return 1;
  amdsmi_status_t s = real(h, type, total);
  if (s != AMDSMI_OK || type != 0 || vgpu_control_disabled() || !total)
    return s;
  int vdev = handle_to_vdev(h);
  if (vdev >= 0) {
    uint64_t lim = vgpu_region_limit(vdev);
    if (lim > 0 && lim < *total) *total = lim;
  }
  return s;
}

amdsmi_status_t amdsmi_get_gpu_memory_usage(amdsmi_processor_handle h,
                                            amdsmi_memory_type_t type,
                                            uint64_t *used) {
  typedef amdsmi_status_t (*fn)(amdsmi_processor_handle, amdsmi_memory_type_t,
                                uint64_t *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_amdsmi("amdsmi_get_gpu_memory_usage");
  if (!real) return 1;
  amdsmi_status_t s = real(h, type, used);
  if (s != AMDSMI_OK || type != 0 || vgpu_control_disabled() || !used)
    return s;
  int vdev = handle_to_vdev(h);
  if (vdev >= 0 && vgpu_region_limit(vdev) > 0)
    *used = vgpu_current_usage(vdev); /* the container's own ledger */
  return s;
}

amdsmi_status_t amdsmi_get_gpu_vram_usage(amdsmi_processor_handle h,
                                          amdsmi_vram_usage_t *info) {
  typedef amdsmi_status_t (*fn)(amdsmi_processor_handle,
                                amdsmi_vram_usage_t *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_amdsmi("amdsmi_get_gpu_vram_usage");
  if (!real) return 1;
  amdsmi_status_t s = real(h, info);
  if (s != AMDSMI_OK || vgpu_control_disabled() || !info) return s;
  int vdev = handle_to_vdev(h);
  if (vdev >= 0) {
    uint64_t lim = vgpu_region_limit(vdev);
    if (lim > 0) {
      uint64_t lim_mb = lim >> 20;
      if (lim_mb < info->vram_total) info->vram_total = (uint32_t)lim_mb;
      info->vram_used = (uint32_t)(vgpu_current_usage(vdev) >> 20);
    }
  }
  return s;
}

typedef struct {
  uint32_t gfx_activity; /* % */
  uint32_t umc_activity;
  uint32_t mm_activity;
  uint32_t reserved[13];
} amdsmi_engine_usage_t; /* amdsmi.h:1147-1152 */

amdsmi_status_t amdsmi_get_gpu_activity(amdsmi_processor_handle h,
                                        amdsmi_engine_usage_t *info) {
  typedef amdsmi_status_t (*fn)(amdsmi_processor_handle,
                                amdsmi_engine_usage_t *);
  static fn real = NULL;
  if (!real) real = (fn)vgpu_real_amdsmi("amdsmi_get_gpu_activity");
  if (!real) return 1;
  amdsmi_status_t s = real(h, info);
  if (s != AMDSMI_OK || vgpu_control_disabled() || !info) return s;
  int vdev = handle_to_vdev(h);
  if (vdev >= 0) {
    uint64_t lim = vgpu_region_sm_limit(vdev);
    if (lim > 0 && lim < 100 && info->gfx_activity > lim)
      info->gfx_activity = (uint32_t)lim; /* quota view, as smi.c does */
  }
  return s;
}
