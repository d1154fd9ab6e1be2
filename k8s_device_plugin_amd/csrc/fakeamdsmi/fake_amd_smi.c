/* Fake libamd_smi for amd-smi spoofing tests (fakehip/fakehsa pattern).
 * Two fake GPUs, 288 GiB each, fixed "physical" usage of 200 GiB. */
#define _GNU_SOURCE
#include <stdint.h>
#include <stdlib.h>

typedef int amdsmi_status_t;
typedef void *amdsmi_processor_handle;
typedef void *amdsmi_socket_handle;
typedef int amdsmi_memory_type_t;
typedef struct {
  uint32_t vram_total;
  uint32_t vram_used;
  uint32_t reserved[2];
} amdsmi_vram_usage_t;

static char g_devs[2]; /* handles = their addresses */

amdsmi_status_t amdsmi_init(uint64_t flags) { (void)flags; return 0; }
amdsmi_status_t amdsmi_shut_down(void) { return 0; }

amdsmi_status_t amdsmi_get_socket_handles(uint32_t *count,
                                          amdsmi_socket_handle *handles) {
  if (count) {
    if (handles && *count >= 1) handles[0] = (void *)0x5;
    *count = 1;
  }
  return 0;
}

amdsmi_status_t amdsmi_get_processor_handles(amdsmi_socket_handle socket,
                                             uint32_t *count,
                                             amdsmi_processor_handle *handles) {
  (void)socket;
  if (!count) return 1;
  if (handles && *count >= 2) {
    handles[0] = &g_devs[0];
    handles[1] = &g_devs[1];
  }
  *count = 2;
  return 0;
}

amdsmi_status_t amdsmi_get_gpu_memory_total(amdsmi_processor_handle h,
                                            amdsmi_memory_type_t type,
                                            uint64_t *total) {
  (void)h;
  if (type != 0 || !total) return 1;
  *total = 288ULL << 30;
  return 0;
}

amdsmi_status_t amdsmi_get_gpu_memory_usage(amdsmi_processor_handle h,
                                            amdsmi_memory_type_t type,
                                            uint64_t *used) {
  (void)h;
  if (type != 0 || !used) return 1;
  *used = 200ULL << 30;
  return 0;
}

amdsmi_status_t amdsmi_get_gpu_vram_usage(amdsmi_processor_handle h,
                                          amdsmi_vram_usage_t *info) {
  (void)h;
  if (!info) return 1;
  info->vram_total = 288u << 10; /* MB */
  info->vram_used = 200u << 10;
  return 0;
}

typedef struct {
  uint32_t gfx_activity;
  uint32_t umc_activity;
  uint32_t mm_activity;
  uint32_t reserved[13];
} amdsmi_engine_usage_t;

amdsmi_status_t amdsmi_get_gpu_activity(amdsmi_processor_handle h,
                                        amdsmi_engine_usage_t *info) {
  (void)h;
  if (!info) return 1;
  info->gfx_activity = 90; /* "physically busy" */
  info->umc_activity = 40;
  info->mm_activity = 0;
  return 0;
}

/* The REAL libamd_smi embeds and exports the whole rsmi_* surface; the
 * interceptor must resolve rsmi symbols to THIS in-process copy via
 * RTLD_NEXT instead of dlopening a second librocm_smi64 (whose C++
 * statics clash -> the round-1 SIGBUS).  The counter proves the embedded
 * copy was the one called. */
static uint64_t g_embedded_rsmi_calls;

uint64_t fake_amdsmi_embedded_rsmi_calls(void) {
  return g_embedded_rsmi_calls;
}

typedef int rsmi_status_t;
typedef int rsmi_memory_type_t;

rsmi_status_t rsmi_dev_memory_total_get(uint32_t dv_ind,
                                        rsmi_memory_type_t type,
                                        uint64_t *total) {
  (void)dv_ind;
  if (type != 0 || !total) return 1;
  g_embedded_rsmi_calls++;
  *total = 288ULL << 30;
  return 0;
}

rsmi_status_t rsmi_dev_memory_usage_get(uint32_t dv_ind,
                                        rsmi_memory_type_t type,
                                        uint64_t *used) {
  (void)dv_ind;
  if (type != 0 || !used) return 1;
  g_embedded_rsmi_calls++;
  *used = 200ULL << 30;
  return 0;
}

rsmi_status_t rsmi_dev_busy_percent_get(uint32_t dv_ind, uint32_t *busy) {
  (void)dv_ind;
  if (!busy) return 1;
  g_embedded_rsmi_calls++;
  *busy = 90;
  return 0;
}
