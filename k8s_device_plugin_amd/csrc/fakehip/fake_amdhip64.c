/* Fake libamdhip64: hardware-free backend for interceptor tests.
 *
 * Pattern borrowed from the reference's test strategy (SURVEY.md §2.4/§4:
 * the cndev C mock driven by env fixtures): a JSON-free fake HIP runtime
 * configured by env vars that the CPU-only CI links/preloads in place of
 * the real library.  Built with SONAME libamdhip64.so so the same test
 * binaries resolve the real runtime on a GPU box and this fake under
 * LD_LIBRARY_PATH here.
 *
 * Env:
 *   FAKE_HIP_DEVICES     device count (default 1)
 *   FAKE_HIP_TOTAL_MEM   bytes per device (default 288 GiB, the MI355X HBM3E)
 */
#define _GNU_SOURCE
#include <pthread.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

typedef int hipError_t;
#define hipSuccess 0
#define hipErrorInvalidValue 1
#define hipErrorOutOfMemory 2

#define MAX_DEV 16
static __thread int t_current = 0;
static uint64_t g_alloc[MAX_DEV];
static uint64_t g_launches;
static uint64_t g_managed_allocs;
static pthread_mutex_t g_mu = PTHREAD_MUTEX_INITIALIZER;

static int dev_count(void) {
  const char *e = getenv("FAKE_HIP_DEVICES");
  int n = e ? atoi(e) : 1;
  return n > 0 && n <= MAX_DEV ? n : 1;
}

static uint64_t total_mem(void) {
  const char *e = getenv("FAKE_HIP_TOTAL_MEM");
  return e ? strtoull(e, NULL, 10) : (288ULL << 30);
}

/* header: 16 bytes before the returned pointer record {size, dev} */
typedef struct { uint64_t size; uint64_t dev; } hdr_t;

hipError_t hipGetDeviceCount(int *count) {
  if (!count) return hipErrorInvalidValue;
  *count = dev_count();
  return hipSuccess;
}

hipError_t hipGetDevice(int *dev) {
  if (!dev) return hipErrorInvalidValue;
  *dev = t_current;
  return hipSuccess;
}

hipError_t hipSetDevice(int dev) {
  if (dev < 0 || dev >= dev_count()) return hipErrorInvalidValue;
  t_current = dev;
  return hipSuccess;
}

static hipError_t fake_alloc(void **ptr, size_t size, int managed) {
  if (!ptr) return hipErrorInvalidValue;
  pthread_mutex_lock(&g_mu);
  /* managed (XNACK/UVM) allocations may exceed device memory — they page
   * to host DRAM; only plain device allocs are bounded by HBM */
  if (!managed && g_alloc[t_current] + size > total_mem()) {
    pthread_mutex_unlock(&g_mu);
    return hipErrorOutOfMemory;
  }
  hdr_t *h = malloc(sizeof(hdr_t) + (size < (1 << 20) ? size : 0) + 16);
  if (!h) {
    pthread_mutex_unlock(&g_mu);
    return hipErrorOutOfMemory;
  }
  h->size = size;
  h->dev = t_current;
  g_alloc[t_current] += size;
  if (managed) g_managed_allocs++;
  pthread_mutex_unlock(&g_mu);
  *ptr = (void *)(h + 1);
  return hipSuccess;
}

hipError_t hipMalloc(void **ptr, size_t size) { return fake_alloc(ptr, size, 0); }

hipError_t hipExtMallocWithFlags(void **ptr, size_t size, unsigned flags) {
  (void)flags;
  return fake_alloc(ptr, size, 0);
}

static unsigned g_last_managed_flags = 0xdeadbeef;

hipError_t hipMallocManaged(void **ptr, size_t size, unsigned flags) {
  __atomic_store_n(&g_last_managed_flags, flags, __ATOMIC_RELAXED);
  return fake_alloc(ptr, size, 1);
}

hipError_t hipMallocAsync(void **ptr, size_t size, void *stream) {
  (void)stream;
  return fake_alloc(ptr, size, 0);
}

hipError_t hipMallocFromPoolAsync(void **ptr, size_t size, void *pool,
                                  void *stream) {
  (void)pool; (void)stream;
  return fake_alloc(ptr, size, 0);
}

hipError_t hipMallocPitch(void **ptr, size_t *pitch, size_t width,
                          size_t height) {
  if (!pitch) return hipErrorInvalidValue;
  size_t p = (width + 255) & ~(size_t)255;
  *pitch = p;
  return fake_alloc(ptr, p * height, 0);
}

hipError_t hipFree(void *ptr) {
  if (!ptr) return hipSuccess;
  hdr_t *h = ((hdr_t *)ptr) - 1;
  pthread_mutex_lock(&g_mu);
  if (g_alloc[h->dev] >= h->size) g_alloc[h->dev] -= h->size;
  pthread_mutex_unlock(&g_mu);
  free(h);
  return hipSuccess;
}

hipError_t hipFreeAsync(void *ptr, void *stream) {
  (void)stream;
  return hipFree(ptr);
}

hipError_t hipMemGetInfo(size_t *free_out, size_t *total_out) {
  pthread_mutex_lock(&g_mu);
  uint64_t used = g_alloc[t_current];
  pthread_mutex_unlock(&g_mu);
  uint64_t total = total_mem();
  if (free_out) *free_out = total > used ? total - used : 0;
  if (total_out) *total_out = total;
  return hipSuccess;
}

hipError_t hipDeviceTotalMem(size_t *bytes, int dev) {
  (void)dev;
  if (bytes) *bytes = total_mem();
  return hipSuccess;
}

/* matches the R0600 prefix the interceptor patches */
typedef struct {
  char name[256];
  char uuid[16];
  char luid[8];
  unsigned luidDeviceNodeMask;
  size_t totalGlobalMem;
  char rest[1024];
} fake_prop_t;

hipError_t hipGetDevicePropertiesR0600(void *prop, int dev) {
  if (!prop || dev < 0 || dev >= dev_count()) return hipErrorInvalidValue;
  fake_prop_t *p = (fake_prop_t *)prop;
  memset(p, 0, sizeof(fake_prop_t));
  snprintf(p->name, sizeof(p->name), "AMD Instinct MI355X (fake)");
  p->totalGlobalMem = total_mem();
  return hipSuccess;
}

hipError_t hipGetDeviceProperties(void *prop, int dev) {
  /* legacy struct: name[256] then totalGlobalMem */
  if (!prop || dev < 0 || dev >= dev_count()) return hipErrorInvalidValue;
  char *c = (char *)prop;
  memset(c, 0, 256 + sizeof(size_t));
  snprintf(c, 256, "AMD Instinct MI355X (fake)");
  uint64_t t = total_mem();
  memcpy(c + 256, &t, sizeof(t));
  return hipSuccess;
}

typedef struct { unsigned x, y, z; } vdim3;

hipError_t hipLaunchKernel(const void *f, vdim3 grid, vdim3 block, void **args,
                           size_t shared, void *stream) {
  (void)f; (void)grid; (void)block; (void)args; (void)shared; (void)stream;
  __atomic_fetch_add(&g_launches, 1, __ATOMIC_RELAXED);
  return hipSuccess;
}

hipError_t hipModuleLaunchKernel(void *f, unsigned gx, unsigned gy, unsigned gz,
                                 unsigned bx, unsigned by, unsigned bz,
                                 unsigned shared, void *stream, void **params,
                                 void **extra) {
  (void)f; (void)gx; (void)gy; (void)gz; (void)bx; (void)by; (void)bz;
  (void)shared; (void)stream; (void)params; (void)extra;
  __atomic_fetch_add(&g_launches, 1, __ATOMIC_RELAXED);
  return hipSuccess;
}

hipError_t hipDeviceSynchronize(void) { return hipSuccess; }
hipError_t hipStreamSynchronize(void *s) { (void)s; return hipSuccess; }

/* ---- graph API subset (real HIP semantics: hipGraphGetNodes returns node
 * handles; hipGraphKernelNodeGetParams fails on non-kernel nodes) -------- */
typedef struct { /* hipKernelNodeParams ABI (hip_runtime_api.h:1492-1499) */
  vdim3 blockDim;
  void **extra;
  void *func;
  vdim3 gridDim;
  void **kernelParams;
  unsigned int sharedMemBytes;
} fake_kernel_node_params_t;

#define FAKE_GRAPH_MAX_NODES 20000
typedef struct {
  int is_kernel;
  fake_kernel_node_params_t params;
} fake_node_t;

typedef struct {
  fake_node_t *nodes;
  size_t n;
} fake_graph_t;

hipError_t hipGraphCreate(void **graph, unsigned int flags) {
  (void)flags;
  if (!graph) return hipErrorInvalidValue;
  fake_graph_t *g = calloc(1, sizeof(fake_graph_t));
  if (!g) return hipErrorOutOfMemory;
  g->nodes = calloc(FAKE_GRAPH_MAX_NODES, sizeof(fake_node_t));
  if (!g->nodes) { free(g); return hipErrorOutOfMemory; }
  *graph = g;
  return hipSuccess;
}

hipError_t hipGraphDestroy(void *graph) {
  fake_graph_t *g = (fake_graph_t *)graph;
  if (g) { free(g->nodes); free(g); }
  return hipSuccess;
}

hipError_t hipGraphAddKernelNode(void **node, void *graph, const void **deps,
                                 size_t ndeps,
                                 const fake_kernel_node_params_t *params) {
  (void)deps; (void)ndeps;
  fake_graph_t *g = (fake_graph_t *)graph;
  if (!g || !params || g->n >= FAKE_GRAPH_MAX_NODES)
    return hipErrorInvalidValue;
  g->nodes[g->n].is_kernel = 1;
  g->nodes[g->n].params = *params;
  if (node) *node = &g->nodes[g->n];
  g->n++;
  return hipSuccess;
}

hipError_t hipGraphAddEmptyNode(void **node, void *graph, const void **deps,
                                size_t ndeps) {
  (void)deps; (void)ndeps;
  fake_graph_t *g = (fake_graph_t *)graph;
  if (!g || g->n >= FAKE_GRAPH_MAX_NODES) return hipErrorInvalidValue;
  g->nodes[g->n].is_kernel = 0;
  if (node) *node = &g->nodes[g->n];
  g->n++;
  return hipSuccess;
}

hipError_t hipGraphGetNodes(void *graph, void **nodes, size_t *numNodes) {
  fake_graph_t *g = (fake_graph_t *)graph;
  if (!g || !numNodes) return hipErrorInvalidValue;
  if (nodes == NULL) {
    *numNodes = g->n;
    return hipSuccess;
  }
  size_t n = *numNodes < g->n ? *numNodes : g->n;
  for (size_t i = 0; i < n; i++) nodes[i] = &g->nodes[i];
  *numNodes = n;
  return hipSuccess;
}

hipError_t hipGraphKernelNodeGetParams(void *node,
                                       fake_kernel_node_params_t *params) {
  fake_node_t *fn = (fake_node_t *)node;
  if (!fn || !params) return hipErrorInvalidValue;
  if (!fn->is_kernel) return hipErrorInvalidValue;
  *params = fn->params;
  return hipSuccess;
}

hipError_t hipGraphInstantiate(void **pGraphExec, void *graph, void *pErrNode,
                               char *pLogBuffer, size_t bufferSize) {
  (void)pErrNode; (void)pLogBuffer; (void)bufferSize;
  if (!pGraphExec || !graph) return hipErrorInvalidValue;
  *pGraphExec = graph; /* exec handle aliases the graph in the fake */
  return hipSuccess;
}

hipError_t hipGraphInstantiateWithFlags(void **pGraphExec, void *graph,
                                        unsigned long long flags) {
  (void)flags;
  return hipGraphInstantiate(pGraphExec, graph, NULL, NULL, 0);
}

hipError_t hipGraphLaunch(void *graphExec, void *stream) {
  (void)stream;
  fake_graph_t *g = (fake_graph_t *)graphExec;
  if (!g) return hipErrorInvalidValue;
  __atomic_fetch_add(&g_launches, g->n, __ATOMIC_RELAXED);
  return hipSuccess;
}

hipError_t hipGraphExecDestroy(void *graphExec) {
  (void)graphExec; /* alias of the graph; freed by hipGraphDestroy */
  return hipSuccess;
}

/* test introspection */
uint64_t fake_hip_launch_count(void) {
  return __atomic_load_n(&g_launches, __ATOMIC_RELAXED);
}
uint64_t fake_hip_managed_count(void) {
  return __atomic_load_n(&g_managed_allocs, __ATOMIC_RELAXED);
}
uint64_t fake_hip_last_managed_flags(void) {
  return __atomic_load_n(&g_last_managed_flags, __ATOMIC_RELAXED);
}
uint64_t fake_hip_device_usage(int dev) {
  pthread_mutex_lock(&g_mu);
  uint64_t u = g_alloc[dev];
  pthread_mutex_unlock(&g_mu);
  return u;
}
