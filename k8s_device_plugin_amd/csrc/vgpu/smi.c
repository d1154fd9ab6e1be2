/* Tool spoofing: rocm-smi / amd-smi inside the container see the vGPU
 * quota, not the physical MI355X.
 *
 * Reference analog: ~260 nvml* exports so nvidia-smi shows quota values
 * (SURVEY.md §2.6 "Tool spoofing").  ROCm design: rocm-smi is a Python CLI
 * that ctypes-dlopens librocm_smi64; our dlopen hook (hook.c) redirects
 * that load to this library, these three functions override the memory and
 * utilization getters, and every other rsmi_* symbol is forwarded to the
 * real library by the dlsym hook — so the whole SMI surface keeps working
 * without re-exporting ~300 symbols.
 *
 * rsmi device indices are NODE-level; the plugin tells us which physical
 * indices belong to this container via VGPU_RSMI_INDICES (comma list, in
 * visible-device order).  Unassigned devices pass through untouched.
 */
#define _GNU_SOURCE
#include "vgpu.h"

#include <dlfcn.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

typedef int rsmi_status_t;
#define RSMI_STATUS_SUCCESS 0
typedef int rsmi_memory_type_t; /* RSMI_MEM_TYPE_VRAM = 0 */

static int rsmi_index_to_vdev(uint32_t dv_ind);

int vgpu_smi_index_to_vdev(uint32_t idx) { return rsmi_index_to_vdev(idx); }

static int rsmi_index_to_vdev(uint32_t dv_ind) {
  static int parsed = 0;
  static int map[VGPU_MAX_DEVICES];
  static int nmap = 0;
  if (!parsed) {
    parsed = 1;
    const char *e = getenv("VGPU_RSMI_INDICES");
    if (e && *e) {
      char tmp[256];
      strncpy(tmp, e, sizeof(tmp) - 1);
      tmp[sizeof(tmp) - 1] = 0;
      char *save = NULL;
      for (char *tok = strtok_r(tmp, ",", &save);
           tok && nmap < VGPU_MAX_DEVICES; tok = strtok_r(NULL, ",", &save))
        map[nmap++] = atoi(tok);
    } else {
      /* containers usually see their devices as 0..n-1 in both worlds */
      vgpu_region_t *r = vgpu_region_get();
      uint64_t n = r ? r->num_devices : 0;
      if (n == 0) n = 1;
      for (int i = 0; i < (int)n && i < VGPU_MAX_DEVICES; i++) map[nmap++] = i;
    }
  }
  for (int i = 0; i < nmap; i++)
    if (map[i] == (int)dv_ind) return i;
  return -1;
}

static void *real_rsmi_sym(const char *name) {
  return vgpu_real_rsmi_sym(name); /* RTLD_NEXT first — ODR-safe (hook.c) */
}

rsmi_status_t rsmi_dev_memory_total_get(uint32_t dv_ind,
                                        rsmi_memory_type_t type,
                                        uint64_t *total) {
  typedef rsmi_status_t (*fn)(uint32_t, rsmi_memory_type_t, uint64_t *);
  static fn real = NULL;
  if (!real) real = (fn)real_rsmi_sym("rsmi_dev_memory_total_get");
  if (!real) return 1;
  rsmi_status_t s = real(dv_ind, type, total);
  if (s != RSMI_STATUS_SUCCESS || type != 0 || vgpu_control_disabled())
    return s;
  int vdev = rsmi_index_to_vdev(dv_ind);
  if (vdev >= 0) {
    uint64_t limit = vgpu_region_limit(vdev);
    if (limit && total) *total = limit;
  }
  return s;
}

rsmi_status_t rsmi_dev_memory_usage_get(uint32_t dv_ind,
                                        rsmi_memory_type_t type,
                                        uint64_t *used) {
  typedef rsmi_status_t (*fn)(uint32_t, rsmi_memory_type_t, uint64_t *);
  static fn real = NULL;
  if (!real) real = (fn)real_rsmi_sym("rsmi_dev_memory_usage_get");
  if (!real) return 1;
  rsmi_status_t s = real(dv_ind, type, used);
  if (s != RSMI_STATUS_SUCCESS || type != 0 || vgpu_control_disabled())
    return s;
  int vdev = rsmi_index_to_vdev(dv_ind);
  if (vdev >= 0) {
    uint64_t limit = vgpu_region_limit(vdev);
    if (limit && used) {
      /* container view: this container's ledger, not host usage */
      uint64_t u = vgpu_current_usage(vdev);
      *used = u > limit ? limit : u;
    }
  }
  return s;
}

rsmi_status_t rsmi_dev_busy_percent_get(uint32_t dv_ind,
                                        uint32_t *busy_percent) {
  typedef rsmi_status_t (*fn)(uint32_t, uint32_t *);
  static fn real = NULL;
  if (!real) real = (fn)real_rsmi_sym("rsmi_dev_busy_percent_get");
  if (!real) return 1;
  rsmi_status_t s = real(dv_ind, busy_percent);
  if (s != RSMI_STATUS_SUCCESS || vgpu_control_disabled()) return s;
  int vdev = rsmi_index_to_vdev(dv_ind);
  if (vdev >= 0 && busy_percent) {
    uint64_t lim = vgpu_region_sm_limit(vdev);
    if (lim > 0 && lim < 100 && *busy_percent > lim)
      *busy_percent = (uint32_t)lim; /* clamp to quota view */
  }
  return s;
}
