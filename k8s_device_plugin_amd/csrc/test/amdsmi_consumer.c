/* PLT-linked amd-smi-style consumer: enumerates processors and prints the
 * memory view per device — run under LD_PRELOAD to verify quota spoofing.
 *
 * AMDSMI_BT=1 installs a SIGBUS/SIGSEGV handler that prints a backtrace
 * before dying (triage tooling for the real-library preload crash,
 * ROUND2_NOTES item 5). */
#define _GNU_SOURCE
#include <dlfcn.h>
#include <execinfo.h>
#include <signal.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <unistd.h>

static void crash_handler(int sig) {
  void *frames[64];
  int n = backtrace(frames, 64);
  dprintf(2, "FATAL signal %d; backtrace (%d frames):\n", sig, n);
  backtrace_symbols_fd(frames, n, 2);
  _exit(128 + sig);
}

typedef int amdsmi_status_t;
typedef void *amdsmi_processor_handle;
typedef void *amdsmi_socket_handle;
typedef struct {
  uint32_t vram_total;
  uint32_t vram_used;
  uint32_t reserved[2];
} amdsmi_vram_usage_t;

extern amdsmi_status_t amdsmi_init(uint64_t);
extern amdsmi_status_t amdsmi_get_socket_handles(uint32_t *,
                                                 amdsmi_socket_handle *);
extern amdsmi_status_t amdsmi_get_processor_handles(amdsmi_socket_handle,
                                                    uint32_t *,
                                                    amdsmi_processor_handle *);
extern amdsmi_status_t amdsmi_get_gpu_memory_total(amdsmi_processor_handle,
                                                   int, uint64_t *);
extern amdsmi_status_t amdsmi_get_gpu_memory_usage(amdsmi_processor_handle,
                                                   int, uint64_t *);
extern amdsmi_status_t amdsmi_get_gpu_vram_usage(amdsmi_processor_handle,
                                                 amdsmi_vram_usage_t *);
typedef struct {
  uint32_t gfx_activity;
  uint32_t umc_activity;
  uint32_t mm_activity;
  uint32_t reserved[13];
} amdsmi_engine_usage_t;
extern amdsmi_status_t amdsmi_get_gpu_activity(amdsmi_processor_handle,
                                               amdsmi_engine_usage_t *);
/* libamd_smi also exports the embedded rsmi surface */
extern int rsmi_dev_memory_total_get(uint32_t, int, uint64_t *);

int main(void) {
  if (getenv("AMDSMI_BT")) {
    signal(SIGBUS, crash_handler);
    signal(SIGSEGV, crash_handler);
  }
  /* AMDSMI_INIT_AMD_GPUS = 1<<1 (amdsmi.h:51; the fake ignores it) */
  amdsmi_init(1 << 1);
  amdsmi_socket_handle sockets[8];
  uint32_t nsock = 8;
  if (amdsmi_get_socket_handles(&nsock, sockets) != 0 || nsock == 0) {
    printf("enumerate failed\n");
    return 1;
  }
  amdsmi_processor_handle hs[16];
  uint32_t n = 16;
  if (amdsmi_get_processor_handles(sockets[0], &n, hs) != 0) {
    printf("enumerate failed\n");
    return 1;
  }
  for (uint32_t i = 0; i < n; i++) {
    uint64_t total = 0, used = 0;
    amdsmi_vram_usage_t vu = {0};
    amdsmi_get_gpu_memory_total(hs[i], 0, &total);
    amdsmi_get_gpu_memory_usage(hs[i], 0, &used);
    amdsmi_get_gpu_vram_usage(hs[i], &vu);
    amdsmi_engine_usage_t act = {0};
    amdsmi_get_gpu_activity(hs[i], &act);
    printf("{\"dev\":%u,\"total\":%llu,\"used\":%llu,"
           "\"vram_total_mb\":%u,\"vram_used_mb\":%u,\"gfx\":%u}\n",
           i, (unsigned long long)total, (unsigned long long)used,
           vu.vram_total, vu.vram_used, act.gfx_activity);
  }
  if (getenv("AMDSMI_CALL_RSMI")) {
    /* exercise the embedded-rsmi interposition path (the one that
     * SIGBUSed in round 1): a PLT call to rsmi_* from a libamd_smi
     * consumer, resolved by the hook via RTLD_NEXT */
    uint64_t total = 0;
    int rc = rsmi_dev_memory_total_get(0, 0, &total);
    long embedded = -1;
    {
      typedef unsigned long long (*cnt_fn)(void);
      /* fake-lib introspection (absent on the real library) */
      void *p = dlsym(RTLD_DEFAULT, "fake_amdsmi_embedded_rsmi_calls");
      if (p) embedded = (long)((cnt_fn)p)();
    }
    printf("{\"rsmi_total\":%llu,\"rsmi_rc\":%d,\"embedded_calls\":%ld}\n",
           (unsigned long long)total, rc, embedded);
  }
  fflush(stdout);
  return 0;
}
