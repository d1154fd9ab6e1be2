/*
 * libvgpu-hip — MI355X in-container vGPU enforcement library.
 *
 * LD_PRELOAD interposer over libamdhip64: hard HBM memory caps, CU-percent
 * launch throttling, device-info virtualization, oversubscription to host
 * DRAM, and rocm-smi spoofing.  MI355X-native re-design of the capabilities
 * of the reference's CUDA/NVML hook (the reference's
 * libvgpu.so source layout is reconstructed at
 * /root/reference/lib/nvidia/libvgpu.so — no code from it is used here).
 *
 * Coordination model: one mmapped shared-memory region per container
 * (path from $VGPU_DEVICE_MEMORY_SHARED_CACHE) shared by
 *   - every hooked process in the container (usage ledger, limits, tokens)
 *   - the node monitor (reads usage, writes blocking/priority feedback).
 * Layout mirrors the reference ABI in spirit with
 * MI355X-appropriate fields (CU tokens for the 256-CU chip).
 */
#ifndef VGPU_H
#define VGPU_H

#include <pthread.h>
#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

#define VGPU_MAGIC 0x4D495655u /* "MIVU" */
/* v3: monitor_scale_* moved AFTER procs[] (appending new fields at the end
 * keeps every pre-existing offset stable across versions) and
 * monitor_interval_ns added.  A version mismatch on attach now REFUSES the
 * region (fail-open, loud) instead of re-initializing it — a v2 library
 * attaching a v3 file must never wipe live accounting. */
#define VGPU_VERSION 3
#define VGPU_MAX_DEVICES 16
#define VGPU_MAX_PROCS 1024
#define VGPU_UUID_LEN 96

/* env vars (set by the device plugin at Allocate; docs/config analog) */
#define ENV_MEM_LIMIT "VGPU_DEVICE_MEMORY_LIMIT"      /* + "_<i>" variants */
#define ENV_CU_LIMIT "VGPU_DEVICE_CU_LIMIT"           /* percent, + _<i>  */
#define ENV_SHARED_CACHE "VGPU_DEVICE_MEMORY_SHARED_CACHE"
#define ENV_OVERSUBSCRIBE "VGPU_OVERSUBSCRIBE"        /* "true" => managed */
#define ENV_TASK_PRIORITY "VGPU_TASK_PRIORITY"        /* 0 high, 1 low */
#define ENV_CORE_POLICY "GPU_CORE_UTILIZATION_POLICY" /* disable|default|force */
#define ENV_DISABLE_CONTROL "VGPU_DISABLE_CONTROL"
#define ENV_ACTIVE_OOM_KILLER "ACTIVE_OOM_KILLER"
#define ENV_LOG_LEVEL "LIBVGPU_LOG_LEVEL"
#define ENV_REAL_HIP "VGPU_REAL_HIP_PATH"             /* test hook: fake lib */
#define ENV_REAL_RSMI "VGPU_REAL_RSMI_PATH"
#define ENV_REAL_HSA "VGPU_REAL_HSA_PATH"
#define ENV_REAL_AMDSMI "VGPU_REAL_AMDSMI_PATH"
#define ENV_DEVICE_UUIDS "VGPU_DEVICE_UUIDS"          /* comma list, monitor correlation */
/* per-process runtime/context overhead charged at region registration
 * (the reference charges CUDA context size the same way, SURVEY §2.6
 * "context/module split"); size string, e.g. "512m"; default 0 */
#define ENV_CONTEXT_OVERHEAD "VGPU_CONTEXT_OVERHEAD"

/* memory accounting split, per device per process (reference ABI:
 * cmd/vGPUmonitor/cudevshr.go:15-58 deviceMemory) */
typedef struct {
  uint64_t context_size; /* runtime/context overhead charged at init */
  uint64_t module_size;  /* code objects (not yet tracked separately) */
  uint64_t buffer_size;  /* explicit allocations */
  uint64_t offset;
  uint64_t total;        /* sum, what the cap checks */
} vgpu_device_memory_t;

typedef struct {
  int32_t pid;      /* in-container pid (pid_ns of the process) */
  int32_t host_pid; /* set by the monitor via cgroup scan; 0 if unknown */
  vgpu_device_memory_t used[VGPU_MAX_DEVICES];
  uint64_t monitor_used[VGPU_MAX_DEVICES]; /* written by monitor (host view) */
  int32_t status;   /* 1 = alive */
  int32_t pad_;
} vgpu_proc_slot_t;

typedef struct {
  uint32_t magic;
  uint32_t version;
  int32_t init_flag;       /* 2 = fully initialized */
  uint32_t owner_pid;      /* pid performing init */
  pthread_mutex_t mutex;   /* process-shared, robust */
  uint64_t num_devices;
  char uuids[VGPU_MAX_DEVICES][VGPU_UUID_LEN];
  uint64_t limit[VGPU_MAX_DEVICES];     /* bytes; 0 = uncapped */
  uint64_t sm_limit[VGPU_MAX_DEVICES];  /* CU percent; 0 or >=100 = no throttle */
  /* CU-throttle token bucket, shared by all procs of the container.
   * Tokens are workgroup-launch credits; refilled by the utilization
   * watcher of whichever process holds the refill lease. */
  int64_t core_tokens[VGPU_MAX_DEVICES];
  int64_t token_fill_rate[VGPU_MAX_DEVICES]; /* tokens/sec, feedback-adjusted */
  uint64_t last_refill_ns;
  vgpu_proc_slot_t procs[VGPU_MAX_PROCS];
  int32_t proc_num;
  int32_t utilization_switch; /* monitor: 1 = enforce CU limit, 0 = free-run */
  int32_t recent_kernel;      /* monitor feedback: <0 = blocked (priority) */
  int32_t priority;           /* this container's task priority */
  uint64_t oversubscribe;     /* 1 = managed-memory alloc mode */
  /* ---- v3 additions (appended; earlier offsets unchanged) ------------- */
  /* Monitor-arbitrated throttle scale (fixed-point x1e6): the node monitor
   * sees EVERY container's region, so it writes one per-device scale to all
   * of them — equal multiplier x entitled CU share = proportional fairness
   * without per-process utilization attribution (which the kernel cannot
   * provide for KFD queues).  The limiter honors it while fresh: younger
   * than 2.5 x monitor_interval_ns (>= 2 s floor). */
  int64_t monitor_scale_fp[VGPU_MAX_DEVICES];
  uint64_t monitor_scale_ts_ns;
  uint64_t monitor_interval_ns; /* written by the monitor; 0 = unknown */
} vgpu_region_t;

/* ---- region API (region.c) ---- */
vgpu_region_t *vgpu_region_get(void);            /* attach/init from env  */
int vgpu_region_lock(vgpu_region_t *r);
void vgpu_region_unlock(vgpu_region_t *r);
uint64_t vgpu_region_device_usage(vgpu_region_t *r, int dev); /* locked sum */
void vgpu_region_add_usage(vgpu_region_t *r, int dev, int64_t delta,
                           int is_context);
uint64_t vgpu_region_limit(int dev);
uint64_t vgpu_region_sm_limit(int dev);
int vgpu_proc_alive(int32_t pid);

/* ---- ledger API (memory.c) ---- */
int vgpu_oom_check(int dev, uint64_t request); /* 0 ok, -1 over limit */
uint64_t vgpu_current_usage(int dev);

/* ---- limiter API (limiter.c) ---- */
void vgpu_limiter_init(void);
void vgpu_limiter_gate(int dev, uint64_t workgroups); /* blocks when throttled */

/* ---- hook core (hook.c) ---- */
void *vgpu_real_hip(const char *sym);   /* resolve real libamdhip64 symbol */
void *vgpu_real_rsmi_handle(void);
/* ODR-safe rsmi resolution: RTLD_NEXT (in-process libamd_smi embedded
 * copy) before dlopening librocm_smi64 — see hook.c */
void *vgpu_real_rsmi_sym(const char *sym);
void *vgpu_real_hsa(const char *sym);   /* resolve real libhsa-runtime64 symbol */
void *vgpu_real_amdsmi(const char *sym); /* resolve real libamd_smi symbol */
int vgpu_smi_index_to_vdev(uint32_t idx); /* smi.c: tool device idx -> vdev */
/* >0 while inside one of our own wrappers: lower-layer hooks (hsa.c) must
 * pass through, not double-count (the HIP runtime allocates via HSA). */
extern __thread int vgpu_tls_passthrough;
int vgpu_initialized(void);
void vgpu_ensure_initialized(void);
int vgpu_control_disabled(void);

/* ---- introspection (region.c): ABI mirror for the Python monitor ---- */
/* Writes a JSON object {field: offset, ..., _size: sizeof(region)} */
int vgpu_region_layout_json(char *buf, size_t buflen);

/* current device helper (device.c) */
int vgpu_current_device(void);

/* logging */
void vgpu_log(int level, const char *fmt, ...);
#define VGPU_ERR 0
#define VGPU_WARN 1
#define VGPU_INFO 2
#define VGPU_DEBUG 3

#ifdef __cplusplus
}
#endif
#endif /* VGPU_H */
