/* Hook core: real-library resolution and dlopen interposition.
 *
 * ROCm interposition model (vs the reference's CUDA one, SURVEY.md §2.6
 * "dlsym bootstrap"): HIP apps link libamdhip64 directly, so plain
 * LD_PRELOAD symbol interposition covers the PLT path; the two escape
 * hatches are (a) dlopen("libamdhip64...") + dlsym(handle, ...), and
 * (b) tools dlopen'ing librocm_smi64 (rocm-smi, amd-smi).  We interpose
 * dlopen and redirect both library names to ourselves, so handle-scoped
 * dlsym finds our hooks; the real libraries are loaded privately via a
 * guard flag that disables the redirect.
 */
#define _GNU_SOURCE
#include "vgpu.h"

#include <dlfcn.h>
#include <pthread.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

static void *g_real_hip = NULL;
static void *g_real_rsmi = NULL;
static pthread_once_t g_init_once = PTHREAD_ONCE_INIT;
static int g_disabled = -1;
static __thread int tls_no_redirect = 0;

typedef void *(*dlopen_fn)(const char *, int);
static dlopen_fn real_dlopen(void) {
  static dlopen_fn fn = NULL;
  if (!fn) fn = (dlopen_fn)dlsym(RTLD_NEXT, "dlopen");
  return fn;
}

int vgpu_control_disabled(void) {
  if (g_disabled < 0)
    g_disabled = getenv(ENV_DISABLE_CONTROL) != NULL;
  return g_disabled;
}

static void *open_real(const char *env_override, const char *const *names,
                       const char *what) {
  tls_no_redirect++;
  void *h = NULL;
  const char *override = env_override ? getenv(env_override) : NULL;
  if (override && *override) {
    h = real_dlopen()(override, RTLD_LAZY | RTLD_LOCAL);
    if (!h)
      vgpu_log(VGPU_ERR, "cannot dlopen %s override %s: %s", what, override,
               dlerror());
  }
  for (int i = 0; !h && names[i]; i++) {
    h = real_dlopen()(names[i], RTLD_LAZY | RTLD_LOCAL);
  }
  tls_no_redirect--;
  if (!h) vgpu_log(VGPU_WARN, "real %s not found", what);
  return h;
}

void *vgpu_real_hip(const char *sym) {
  if (!g_real_hip) {
    static const char *const names[] = {"libamdhip64.so.7", "libamdhip64.so.6",
                                        "libamdhip64.so", NULL};
    g_real_hip = open_real(ENV_REAL_HIP, names, "libamdhip64");
  }
  if (!g_real_hip) return NULL;
  void *p = dlsym(g_real_hip, sym);
  if (!p) vgpu_log(VGPU_DEBUG, "real hip symbol %s not found", sym);
  return p;
}

void *vgpu_real_rsmi_handle(void) {
  if (!g_real_rsmi) {
    static const char *const names[] = {"librocm_smi64.so.7",
                                        "librocm_smi64.so.6",
                                        "librocm_smi64.so", NULL};
    g_real_rsmi = open_real(ENV_REAL_RSMI, names, "librocm_smi64");
  }
  return g_real_rsmi;
}

/* rsmi symbol resolution, ODR-safe order (amd-smi crash triage,
 * gpurun_out/amdsmi_triage_r2.log): libamd_smi EMBEDS the amd::smi C++
 * classes that librocm_smi64 also defines, with layouts that differ
 * between the two builds — dlopening librocm_smi64 into a process that
 * already has libamd_smi loaded makes librocm_smi's static initializers
 * bind to libamd_smi's incompatible copies and SIGBUS.  So: resolve
 * through RTLD_NEXT first (the implementation already in the process —
 * libamd_smi exports the whole rsmi_* surface), and only dlopen the real
 * librocm_smi64 when the process has none (the rocm-smi CLI dlopen path,
 * where no libamd_smi is resident and the load is safe). */
void *vgpu_real_rsmi_sym(const char *sym) {
  if (!getenv(ENV_REAL_RSMI)) { /* test override must stay authoritative */
    void *p = dlsym(RTLD_NEXT, sym);
    if (p) return p;
  }
  void *h = vgpu_real_rsmi_handle();
  return h ? dlsym(h, sym) : NULL;
}

__thread int vgpu_tls_passthrough = 0;

static void *g_real_amdsmi = NULL;

void *vgpu_real_amdsmi(const char *sym) {
  /* same ODR hazard in reverse (librocm_smi64 resident, then dlopen
   * libamd_smi): prefer the copy already linked into the process */
  if (!getenv(ENV_REAL_AMDSMI)) {
    void *p = dlsym(RTLD_NEXT, sym);
    if (p) return p;
  }
  if (!g_real_amdsmi) {
    static const char *const names[] = {"libamd_smi.so.26", "libamd_smi.so.25",
                                        "libamd_smi.so", NULL};
    g_real_amdsmi = open_real(ENV_REAL_AMDSMI, names, "libamd_smi");
  }
  if (!g_real_amdsmi) return NULL;
  return dlsym(g_real_amdsmi, sym);
}

static void *g_real_hsa = NULL;

void *vgpu_real_hsa(const char *sym) {
  if (!g_real_hsa) {
    static const char *const names[] = {"libhsa-runtime64.so.1",
                                        "libhsa-runtime64.so", NULL};
    g_real_hsa = open_real(ENV_REAL_HSA, names, "libhsa-runtime64");
  }
  if (!g_real_hsa) return NULL;
  return dlsym(g_real_hsa, sym);
}

static void do_init(void) {
  if (vgpu_control_disabled()) {
    vgpu_log(VGPU_INFO, "control disabled via %s", ENV_DISABLE_CONTROL);
    return;
  }
  vgpu_region_t *r = vgpu_region_get();
  if (!r) {
    vgpu_log(VGPU_ERR, "shared region unavailable; enforcement off");
    g_disabled = 1;
    return;
  }
  vgpu_limiter_init();
  vgpu_log(VGPU_INFO, "libvgpu-hip initialized (devices=%llu)",
           (unsigned long long)r->num_devices);
}

void vgpu_ensure_initialized(void) { pthread_once(&g_init_once, do_init); }

int vgpu_initialized(void) { return !vgpu_control_disabled(); }

/* ---- dlopen interposition ------------------------------------------- */
#ifndef VGPU_NO_DLOPEN
static const char *self_path(void) {
  static char path[4096];
  if (!path[0]) {
    Dl_info info;
    if (dladdr((void *)self_path, &info) && info.dli_fname)
      snprintf(path, sizeof(path), "%s", info.dli_fname);
  }
  return path[0] ? path : NULL;
}

static void *g_self_handle = NULL; /* handle apps got from a redirect */

void *dlopen(const char *filename, int flags) {
  if (filename && !tls_no_redirect && !vgpu_control_disabled()) {
    if (strstr(filename, "libamdhip64") || strstr(filename, "librocm_smi64") ||
        strstr(filename, "libhsa-runtime64") || strstr(filename, "libamd_smi")) {
      const char *self = self_path();
      if (self) {
        vgpu_log(VGPU_INFO, "redirecting dlopen(%s) to %s", filename, self);
        void *h = real_dlopen()(self, flags);
        if (h) {
          g_self_handle = h;
          /* the real library is resolved LAZILY on the first forwarded
           * miss — an eager load here would force librocm_smi64 and
           * libamd_smi into one process even when the app never calls
           * into the other, re-creating the ODR crash the RTLD_NEXT
           * resolution order avoids */
          return h;
        }
      }
    }
  }
  return real_dlopen()(filename, flags);
}
#endif /* VGPU_NO_DLOPEN */

/* dlsym interposition: a redirected handle must still resolve the hundreds
 * of symbols we do not hook — forward misses to the real library, chosen by
 * symbol prefix. */
#ifndef VGPU_NO_DLSYM
typedef void *(*dlsym_fn)(void *, const char *);
static dlsym_fn real_dlsym(void) {
  static dlsym_fn fn = NULL;
  if (!fn) {
    fn = (dlsym_fn)dlvsym(RTLD_NEXT, "dlsym", "GLIBC_2.2.5");
    if (!fn) fn = (dlsym_fn)dlvsym(RTLD_NEXT, "dlsym", "GLIBC_2.34");
    if (!fn) {
      /* last resort: libdl/libc direct */
      void *h = real_dlopen()("libdl.so.2", RTLD_LAZY | RTLD_LOCAL);
      if (h) fn = (dlsym_fn)dlvsym(h, "dlsym", "GLIBC_2.2.5");
    }
  }
  return fn;
}

void *dlsym(void *handle, const char *symbol) {
  dlsym_fn real = real_dlsym();
  if (!real) return NULL;
  void *p = real(handle, symbol);
  /* volatile copy: glibc marks symbol __nonnull, but defensive NULL
   * handling matters for an interposer (-Wnonnull-compare otherwise) */
  const char *volatile sym_v = symbol;
  if (!p && sym_v != NULL && g_self_handle && handle == g_self_handle) {
    tls_no_redirect++;
    if (strncmp(symbol, "rsmi_", 5) == 0) {
      p = vgpu_real_rsmi_sym(symbol);
    } else if (strncmp(symbol, "amdsmi_", 7) == 0) {
      p = vgpu_real_amdsmi(symbol);
    } else if (strncmp(symbol, "hsa_", 4) == 0) {
      p = vgpu_real_hsa(symbol);
    } else {
      p = vgpu_real_hip(symbol);
    }
    tls_no_redirect--;
  }
  return p;
}
#endif /* VGPU_NO_DLSYM */
