/* Shared-memory region: one mmapped file per container.
 *
 * Creation is raced-safe via flock on the region file; cross-process mutual
 * exclusion afterwards is a process-shared ROBUST pthread mutex stored in
 * the region (dead-owner recovery replaces the reference's semaphore +
 * owner-pid timeout repair, SURVEY.md §2.6 "Multiprocess shared region").
 * The Python monitor does word-sized reads/writes only (feedback fields),
 * which need no lock.
 */
#define _GNU_SOURCE
#include "vgpu.h"

#include <errno.h>
#include <fcntl.h>
#include <limits.h>
#include <signal.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/file.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <time.h>
#include <unistd.h>

static vgpu_region_t *g_region = NULL;
static pthread_once_t g_region_once = PTHREAD_ONCE_INIT;
static int g_proc_slot = -1;

static uint64_t parse_size(const char *s) {
  /* "73728m", "72g", "288G", plain bytes */
  char *end = NULL;
  double v = strtod(s, &end);
  if (end == s) return 0;
  while (*end == ' ') end++;
  switch (*end) {
    case 'k': case 'K': return (uint64_t)(v * 1024.0);
    case 'm': case 'M': return (uint64_t)(v * 1024.0 * 1024.0);
    case 'g': case 'G': return (uint64_t)(v * 1024.0 * 1024.0 * 1024.0);
    case 't': case 'T': return (uint64_t)(v * 1024.0 * 1024.0 * 1024.0 * 1024.0);
    default: return (uint64_t)v;
  }
}

static uint64_t env_limit_for(const char *base, int dev, int as_size) {
  char key[128];
  snprintf(key, sizeof(key), "%s_%d", base, dev);
  const char *e = getenv(key);
  if (!e) e = getenv(base);
  if (!e) return 0;
  return as_size ? parse_size(e) : (uint64_t)strtoull(e, NULL, 10);
}

int vgpu_proc_alive(int32_t pid) {
  if (pid <= 0) return 0;
  return kill(pid, 0) == 0 || errno == EPERM;
}

static void region_init_fields(vgpu_region_t *r) {
  memset(r, 0, sizeof(*r));
  r->magic = VGPU_MAGIC;
  r->version = VGPU_VERSION;
  r->owner_pid = (uint32_t)getpid();

  pthread_mutexattr_t attr;
  pthread_mutexattr_init(&attr);
  pthread_mutexattr_setpshared(&attr, PTHREAD_PROCESS_SHARED);
  pthread_mutexattr_setrobust(&attr, PTHREAD_MUTEX_ROBUST);
  pthread_mutex_init(&r->mutex, &attr);
  pthread_mutexattr_destroy(&attr);

  for (int i = 0; i < VGPU_MAX_DEVICES; i++) {
    r->limit[i] = env_limit_for(ENV_MEM_LIMIT, i, 1);
    r->sm_limit[i] = env_limit_for(ENV_CU_LIMIT, i, 0);
    /* start the bucket generous: one second of fill at the initial rate */
    r->token_fill_rate[i] = 0; /* limiter sets on first refill */
    r->core_tokens[i] = 0;
  }
  const char *uuids = getenv(ENV_DEVICE_UUIDS);
  int n = 0;
  if (uuids) {
    char tmp[VGPU_MAX_DEVICES * VGPU_UUID_LEN];
    strncpy(tmp, uuids, sizeof(tmp) - 1);
    tmp[sizeof(tmp) - 1] = 0;
    char *save = NULL;
    for (char *tok = strtok_r(tmp, ",", &save); tok && n < VGPU_MAX_DEVICES;
         tok = strtok_r(NULL, ",", &save)) {
      strncpy(r->uuids[n], tok, VGPU_UUID_LEN - 1);
      n++;
    }
  }
  r->num_devices = n;
  const char *prio = getenv(ENV_TASK_PRIORITY);
  r->priority = prio ? atoi(prio) : 1;
  const char *over = getenv(ENV_OVERSUBSCRIBE);
  r->oversubscribe = (over && strcasecmp(over, "true") == 0) ? 1 : 0;
  r->utilization_switch = 1;
  r->recent_kernel = 1;
  __atomic_store_n(&r->init_flag, 2, __ATOMIC_RELEASE);
}

static void region_register_proc(vgpu_region_t *r) {
  vgpu_region_lock(r);
  int32_t me = (int32_t)getpid();
  int free_slot = -1;
  for (int i = 0; i < VGPU_MAX_PROCS; i++) {
    vgpu_proc_slot_t *s = &r->procs[i];
    if (s->pid == me) { free_slot = i; break; }
    if (free_slot < 0 && (s->pid == 0 || !vgpu_proc_alive(s->pid)))
      free_slot = i;
  }
  if (free_slot >= 0) {
    vgpu_proc_slot_t *s = &r->procs[free_slot];
    if (s->pid != me) memset(s, 0, sizeof(*s));
    s->pid = me;
    s->status = 1;
    g_proc_slot = free_slot;
    /* charge the configured per-process context overhead on every device
     * this container sees, so quota math covers runtime reservations */
    const char *ctx = getenv(ENV_CONTEXT_OVERHEAD);
    if (ctx && *ctx) {
      uint64_t bytes = parse_size(ctx);
      uint64_t n = r->num_devices ? r->num_devices : 1;
      for (uint64_t d = 0; d < n && d < VGPU_MAX_DEVICES; d++) {
        s->used[d].context_size = bytes;
        s->used[d].total = s->used[d].context_size +
                           s->used[d].module_size + s->used[d].buffer_size;
      }
    }
    int cnt = 0;
    for (int i = 0; i < VGPU_MAX_PROCS; i++)
      if (r->procs[i].pid && vgpu_proc_alive(r->procs[i].pid)) cnt++;
    r->proc_num = cnt;
  } else {
    vgpu_log(VGPU_ERR, "no free proc slot in shared region");
  }
  vgpu_region_unlock(r);
}

static void atfork_child(void);

static void region_open(void) {
  const char *path = getenv(ENV_SHARED_CACHE);
  char fallback[PATH_MAX];
  if (!path || !*path) {
    /* reference fallback: /tmp/cudevshr.cache */
    snprintf(fallback, sizeof(fallback), "/tmp/vgpu-mi355x.cache");
    path = fallback;
  }
  int fd = open(path, O_CREAT | O_RDWR, 0666);
  if (fd < 0) {
    vgpu_log(VGPU_ERR, "cannot open shared region %s: %s", path, strerror(errno));
    return;
  }
  if (flock(fd, LOCK_EX) != 0)
    vgpu_log(VGPU_WARN, "flock %s failed: %s", path, strerror(errno));

  struct stat st;
  fstat(fd, &st);
  /* Version safety BEFORE any ftruncate: a file that already carries our
   * magic but a different version (mixed library versions in a container,
   * or an old region file surviving a library upgrade) must not be
   * re-initialized — that would wipe live processes' accounting — and must
   * not be read through mismatched offsets.  Refuse loudly; enforcement
   * for THIS process is then off (fail-open, matching hook-less behavior),
   * which is observable in the logs rather than silently corrupting. */
  if (st.st_size >= (off_t)(2 * sizeof(uint32_t))) {
    uint32_t head[2] = {0, 0};
    if (pread(fd, head, sizeof(head), 0) == (ssize_t)sizeof(head) &&
        head[0] == VGPU_MAGIC && head[1] != VGPU_VERSION) {
      vgpu_log(VGPU_ERR,
               "shared region %s has version %u but this library is v%u; "
               "refusing to attach (enforcement disabled for this process; "
               "redeploy with matching library versions)",
               path, head[1], VGPU_VERSION);
      flock(fd, LOCK_UN);
      close(fd);
      return;
    }
  }
  int fresh = st.st_size < (off_t)sizeof(vgpu_region_t);
  if (fresh && ftruncate(fd, sizeof(vgpu_region_t)) != 0) {
    vgpu_log(VGPU_ERR, "ftruncate %s failed: %s", path, strerror(errno));
    flock(fd, LOCK_UN);
    close(fd);
    return;
  }
  void *mem = mmap(NULL, sizeof(vgpu_region_t), PROT_READ | PROT_WRITE,
                   MAP_SHARED, fd, 0);
  if (mem == MAP_FAILED) {
    vgpu_log(VGPU_ERR, "mmap %s failed: %s", path, strerror(errno));
    flock(fd, LOCK_UN);
    close(fd);
    return;
  }
  vgpu_region_t *r = (vgpu_region_t *)mem;
  if (fresh || r->magic != VGPU_MAGIC ||
      __atomic_load_n(&r->init_flag, __ATOMIC_ACQUIRE) != 2) {
    region_init_fields(r);
    vgpu_log(VGPU_INFO, "initialized shared region %s (limit0=%llu MiB sm0=%llu%%)",
             path, (unsigned long long)(r->limit[0] >> 20),
             (unsigned long long)r->sm_limit[0]);
  }
  flock(fd, LOCK_UN);
  close(fd); /* mapping persists */
  g_region = r;
  region_register_proc(r);
  pthread_atfork(NULL, NULL, atfork_child);
}

/* fork() support: the child inherits the mapping but NOT the proc slot —
 * its pid differs, so its allocations must go to its own slot (PyTorch
 * dataloader workers etc.).  Registered once via pthread_atfork
 * (forward-declared above region_open). */
static void atfork_child(void) {
  if (g_region) {
    g_proc_slot = -1;
    region_register_proc(g_region);
  }
}

vgpu_region_t *vgpu_region_get(void) {
  pthread_once(&g_region_once, region_open);
  return g_region;
}

int vgpu_region_lock(vgpu_region_t *r) {
  struct timespec ts;
  clock_gettime(CLOCK_REALTIME, &ts);
  ts.tv_sec += 5;
  int rc = pthread_mutex_timedlock(&r->mutex, &ts);
  if (rc == EOWNERDEAD) {
    vgpu_log(VGPU_WARN, "shared region lock owner died; repairing");
    pthread_mutex_consistent(&r->mutex);
    rc = 0;
  } else if (rc == ETIMEDOUT) {
    /* Stuck live owner or corrupted mutex: re-init (the reference's
     * fix_lock_shrreg timeout repair). */
    vgpu_log(VGPU_ERR, "shared region lock timeout; re-initializing mutex");
    pthread_mutexattr_t attr;
    pthread_mutexattr_init(&attr);
    pthread_mutexattr_setpshared(&attr, PTHREAD_PROCESS_SHARED);
    pthread_mutexattr_setrobust(&attr, PTHREAD_MUTEX_ROBUST);
    pthread_mutex_init(&r->mutex, &attr);
    pthread_mutexattr_destroy(&attr);
    rc = pthread_mutex_lock(&r->mutex);
  }
  return rc;
}

void vgpu_region_unlock(vgpu_region_t *r) { pthread_mutex_unlock(&r->mutex); }

uint64_t vgpu_region_device_usage(vgpu_region_t *r, int dev) {
  /* caller holds the lock; prunes dead processes as it sums
   * (reference rm_quitted_process) */
  uint64_t sum = 0;
  for (int i = 0; i < VGPU_MAX_PROCS; i++) {
    vgpu_proc_slot_t *s = &r->procs[i];
    if (s->pid == 0) continue;
    if (!vgpu_proc_alive(s->pid)) {
      memset(s, 0, sizeof(*s));
      continue;
    }
    sum += s->used[dev].total;
  }
  return sum;
}

void vgpu_region_add_usage(vgpu_region_t *r, int dev, int64_t delta,
                           int is_context) {
  if (g_proc_slot < 0 || dev < 0 || dev >= VGPU_MAX_DEVICES) return;
  vgpu_region_lock(r);
  vgpu_proc_slot_t *s = &r->procs[g_proc_slot];
  uint64_t *bucket = is_context ? &s->used[dev].context_size
                                : &s->used[dev].buffer_size;
  if (delta < 0 && (uint64_t)(-delta) > *bucket)
    *bucket = 0;
  else
    *bucket += delta;
  s->used[dev].total = s->used[dev].context_size + s->used[dev].module_size +
                       s->used[dev].buffer_size;
  vgpu_region_unlock(r);
}

uint64_t vgpu_region_limit(int dev) {
  vgpu_region_t *r = vgpu_region_get();
  if (!r || dev < 0 || dev >= VGPU_MAX_DEVICES) return 0;
  return r->limit[dev];
}

uint64_t vgpu_region_sm_limit(int dev) {
  vgpu_region_t *r = vgpu_region_get();
  if (!r || dev < 0 || dev >= VGPU_MAX_DEVICES) return 0;
  return r->sm_limit[dev];
}

static void __attribute__((destructor)) region_cleanup(void) {
  vgpu_region_t *r = g_region;
  if (!r || g_proc_slot < 0) return;
  /* best-effort slot release; a crash leaves it to the liveness pruning */
  vgpu_proc_slot_t *s = &r->procs[g_proc_slot];
  if (s->pid == (int32_t)getpid()) memset(s, 0, sizeof(*s));
}

#define OFF(f) ((unsigned long)offsetof(vgpu_region_t, f))
int vgpu_region_layout_json(char *buf, size_t buflen) {
  return snprintf(
      buf, buflen,
      "{\"_size\":%lu,\"magic\":%lu,\"version\":%lu,\"init_flag\":%lu,"
      "\"owner_pid\":%lu,\"num_devices\":%lu,\"uuids\":%lu,\"limit\":%lu,"
      "\"sm_limit\":%lu,\"core_tokens\":%lu,\"token_fill_rate\":%lu,"
      "\"last_refill_ns\":%lu,\"monitor_scale_fp\":%lu,"
      "\"monitor_scale_ts_ns\":%lu,\"monitor_interval_ns\":%lu,"
      "\"procs\":%lu,\"proc_num\":%lu,"
      "\"utilization_switch\":%lu,\"recent_kernel\":%lu,\"priority\":%lu,"
      "\"oversubscribe\":%lu,\"_proc_slot_size\":%lu,\"_proc_pid\":%lu,"
      "\"_proc_host_pid\":%lu,\"_proc_used\":%lu,\"_proc_monitor_used\":%lu,"
      "\"_proc_status\":%lu,\"_devmem_size\":%lu,\"_max_devices\":%d,"
      "\"_max_procs\":%d,\"_uuid_len\":%d}",
      (unsigned long)sizeof(vgpu_region_t), OFF(magic), OFF(version),
      OFF(init_flag), OFF(owner_pid), OFF(num_devices), OFF(uuids),
      OFF(limit), OFF(sm_limit), OFF(core_tokens), OFF(token_fill_rate),
      OFF(last_refill_ns), OFF(monitor_scale_fp), OFF(monitor_scale_ts_ns),
      OFF(monitor_interval_ns), OFF(procs), OFF(proc_num),
      OFF(utilization_switch), OFF(recent_kernel), OFF(priority),
      OFF(oversubscribe), (unsigned long)sizeof(vgpu_proc_slot_t),
      (unsigned long)offsetof(vgpu_proc_slot_t, pid),
      (unsigned long)offsetof(vgpu_proc_slot_t, host_pid),
      (unsigned long)offsetof(vgpu_proc_slot_t, used),
      (unsigned long)offsetof(vgpu_proc_slot_t, monitor_used),
      (unsigned long)offsetof(vgpu_proc_slot_t, status),
      (unsigned long)sizeof(vgpu_device_memory_t), VGPU_MAX_DEVICES,
      VGPU_MAX_PROCS, VGPU_UUID_LEN);
}
