/* hip_consumer — scriptable test binary exercising HIP through the PLT.
 *
 * Linked against SONAME libamdhip64.so, so:
 *   - CPU CI: LD_LIBRARY_PATH points at the fake runtime, LD_PRELOAD at
 *     libvgpu-hip.so -> true preload interposition is what's under test;
 *   - GPU box: the same binary resolves the real ROCm runtime.
 *
 * Commands (argv, executed in order), each printing one JSON line:
 *   alloc <bytes>        hipMalloc; remembers the pointer on a stack
 *   allocmanaged <bytes> hipMallocManaged
 *   free                 hipFree the most recent pointer
 *   meminfo              hipMemGetInfo
 *   totalmem             hipDeviceTotalMem(dev 0)
 *   launch <n> <grid>    n hipLaunchKernel calls with grid workgroups
 *   graphlaunch <launches> <nodes> <grid>
 *                        build a graph of <nodes> kernel nodes (grid wgs
 *                        each), instantiate, replay it <launches> times
 *   sleep <ms>
 *   setdevice <i>
 */
#define _GNU_SOURCE
#include <dlfcn.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/wait.h>
#include <unistd.h>
#include <time.h>

typedef int hipError_t;
typedef struct { unsigned x, y, z; } vdim3;

extern hipError_t hipMalloc(void **, size_t);
extern hipError_t hipMallocManaged(void **, size_t, unsigned);
extern hipError_t hipFree(void *);
extern hipError_t hipMemGetInfo(size_t *, size_t *);
extern hipError_t hipDeviceTotalMem(size_t *, int);
extern hipError_t hipSetDevice(int);
extern hipError_t hipLaunchKernel(const void *, vdim3, vdim3, void **, size_t,
                                  void *);

typedef struct { /* hipKernelNodeParams ABI */
  vdim3 blockDim;
  void **extra;
  void *func;
  vdim3 gridDim;
  void **kernelParams;
  unsigned int sharedMemBytes;
} knode_params_t;

extern hipError_t hipGraphCreate(void **, unsigned);
extern hipError_t hipGraphDestroy(void *);
extern hipError_t hipGraphAddKernelNode(void **, void *, const void **, size_t,
                                        const knode_params_t *);
extern hipError_t hipGraphInstantiate(void **, void *, void *, char *, size_t);
extern hipError_t hipGraphLaunch(void *, void *);
extern hipError_t hipGraphExecDestroy(void *);

static double now_s(void) {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec + ts.tv_nsec * 1e-9;
}

int main(int argc, char **argv) {
  void *stack[1024];
  int sp = 0;
  for (int i = 1; i < argc; i++) {
    const char *cmd = argv[i];
    if (strcmp(cmd, "alloc") == 0 && i + 1 < argc) {
      size_t n = strtoull(argv[++i], NULL, 10);
      void *p = NULL;
      hipError_t e = hipMalloc(&p, n);
      if (e == 0 && sp < 1024) stack[sp++] = p;
      printf("{\"cmd\":\"alloc\",\"bytes\":%zu,\"err\":%d}\n", n, e);
    } else if (strcmp(cmd, "allocmanaged") == 0 && i + 1 < argc) {
      size_t n = strtoull(argv[++i], NULL, 10);
      void *p = NULL;
      hipError_t e = hipMallocManaged(&p, n, 1u);
      if (e == 0 && sp < 1024) stack[sp++] = p;
      printf("{\"cmd\":\"allocmanaged\",\"bytes\":%zu,\"err\":%d}\n", n, e);
    } else if (strcmp(cmd, "free") == 0) {
      hipError_t e = 1;
      if (sp > 0) e = hipFree(stack[--sp]);
      printf("{\"cmd\":\"free\",\"err\":%d}\n", e);
    } else if (strcmp(cmd, "meminfo") == 0) {
      size_t f = 0, t = 0;
      hipError_t e = hipMemGetInfo(&f, &t);
      printf("{\"cmd\":\"meminfo\",\"free\":%zu,\"total\":%zu,\"err\":%d}\n",
             f, t, e);
    } else if (strcmp(cmd, "totalmem") == 0) {
      size_t t = 0;
      hipError_t e = hipDeviceTotalMem(&t, 0);
      printf("{\"cmd\":\"totalmem\",\"total\":%zu,\"err\":%d}\n", t, e);
    } else if (strcmp(cmd, "setdevice") == 0 && i + 1 < argc) {
      int d = atoi(argv[++i]);
      printf("{\"cmd\":\"setdevice\",\"dev\":%d,\"err\":%d}\n", d,
             hipSetDevice(d));
    } else if (strcmp(cmd, "launch") == 0 && i + 2 < argc) {
      long n = atol(argv[++i]);
      unsigned grid = (unsigned)atoi(argv[++i]);
      vdim3 g = {grid, 1, 1}, b = {64, 1, 1};
      double t0 = now_s();
      hipError_t e = 0;
      for (long k = 0; k < n && e == 0; k++)
        e = hipLaunchKernel((void *)main, g, b, NULL, 0, NULL);
      double dt = now_s() - t0;
      printf(
          "{\"cmd\":\"launch\",\"n\":%ld,\"grid\":%u,\"seconds\":%.6f,"
          "\"err\":%d}\n",
          n, grid, dt, e);
    } else if (strcmp(cmd, "launchb") == 0 && i + 3 < argc) {
      /* like launch, with an explicit block size (wavefront-cost tests) */
      long n = atol(argv[++i]);
      unsigned grid = (unsigned)atoi(argv[++i]);
      unsigned block = (unsigned)atoi(argv[++i]);
      vdim3 g = {grid, 1, 1}, b = {block, 1, 1};
      double t0 = now_s();
      hipError_t e = 0;
      for (long k = 0; k < n && e == 0; k++)
        e = hipLaunchKernel((void *)main, g, b, NULL, 0, NULL);
      double dt = now_s() - t0;
      printf(
          "{\"cmd\":\"launchb\",\"n\":%ld,\"grid\":%u,\"block\":%u,"
          "\"seconds\":%.6f,\"err\":%d}\n",
          n, grid, block, dt, e);
    } else if (strcmp(cmd, "graphlaunch") == 0 && i + 3 < argc) {
      long launches = atol(argv[++i]);
      long nodes = atol(argv[++i]);
      unsigned grid = (unsigned)atoi(argv[++i]);
      void *graph = NULL, *exec = NULL;
      hipError_t e = hipGraphCreate(&graph, 0);
      knode_params_t p;
      memset(&p, 0, sizeof(p));
      p.func = (void *)main;
      p.gridDim.x = grid;
      p.gridDim.y = p.gridDim.z = 1;
      p.blockDim.x = 64;
      p.blockDim.y = p.blockDim.z = 1;
      for (long k = 0; k < nodes && e == 0; k++)
        e = hipGraphAddKernelNode(NULL, graph, NULL, 0, &p);
      if (e == 0) e = hipGraphInstantiate(&exec, graph, NULL, NULL, 0);
      double t0 = now_s();
      for (long k = 0; k < launches && e == 0; k++)
        e = hipGraphLaunch(exec, NULL);
      double dt = now_s() - t0;
      if (exec) hipGraphExecDestroy(exec);
      if (graph) hipGraphDestroy(graph);
      printf(
          "{\"cmd\":\"graphlaunch\",\"launches\":%ld,\"nodes\":%ld,"
          "\"grid\":%u,\"seconds\":%.6f,\"err\":%d}\n",
          launches, nodes, grid, dt, e);
    } else if (strcmp(cmd, "stats") == 0) {
      /* fake-runtime introspection; -1 on the real runtime */
      long launches = -1, managed = -1;
      typedef unsigned long long (*cnt_fn)(void);
      long mflags = -1;
      cnt_fn lf = (cnt_fn)dlsym(RTLD_DEFAULT, "fake_hip_launch_count");
      cnt_fn mf = (cnt_fn)dlsym(RTLD_DEFAULT, "fake_hip_managed_count");
      cnt_fn ff = (cnt_fn)dlsym(RTLD_DEFAULT, "fake_hip_last_managed_flags");
      if (lf) launches = (long)lf();
      if (mf) managed = (long)mf();
      if (ff) mflags = (long)ff();
      printf(
          "{\"cmd\":\"stats\",\"launches\":%ld,\"managed\":%ld,"
          "\"managed_flags\":%ld}\n",
          launches, managed, mflags);
    } else if (strcmp(cmd, "forkhold") == 0 && i + 2 < argc) {
      /* fork(): the child's allocations must account to ITS OWN proc slot
       * (atfork re-registration) and vanish when it exits */
      size_t n = strtoull(argv[++i], NULL, 10);
      long ms = atol(argv[++i]);
      pid_t pid = fork();
      if (pid == 0) {
        void *p = NULL;
        hipError_t e = hipMalloc(&p, n);
        struct timespec ts;
        ts.tv_sec = ms / 1000;
        ts.tv_nsec = (ms % 1000) * 1000000L;
        nanosleep(&ts, NULL);
        _exit(e == 0 ? 0 : 42);
      }
      struct timespec w = {0, 300 * 1000000L};
      nanosleep(&w, NULL);
      size_t f = 0, t = 0;
      hipMemGetInfo(&f, &t);
      printf("{\"cmd\":\"forkhold\",\"free\":%zu,\"total\":%zu}\n", f, t);
      fflush(stdout);
      int st = 0;
      waitpid(pid, &st, 0);
      printf("{\"cmd\":\"forkdone\",\"status\":%d}\n", WEXITSTATUS(st));
    } else if (strcmp(cmd, "sleep") == 0 && i + 1 < argc) {
      struct timespec ts;
      long ms = atol(argv[++i]);
      ts.tv_sec = ms / 1000;
      ts.tv_nsec = (ms % 1000) * 1000000L;
      nanosleep(&ts, NULL);
      printf("{\"cmd\":\"sleep\",\"ms\":%ld}\n", ms);
    } else {
      fprintf(stderr, "unknown command %s\n", cmd);
      return 2;
    }
    fflush(stdout);
  }
  return 0;
}
