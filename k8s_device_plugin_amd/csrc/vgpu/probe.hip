/* libvgpu_probe — minimal gfx950 device-code companion to the interceptor.
 *
 * Purposes:
 *  - GPU numerics check that the interposed path still computes (vecadd);
 *  - controllable load generator for CU-throttle tests (burn / storm):
 *    the burn kernel spins on s_memtime-equivalent clock64() so busy% is
 *    deterministic regardless of memory behavior;
 *  - evidence that in-tree native code actually runs on the box.
 *
 * All entry points are extern "C" for ctypes.  64-wide wavefronts; grids are
 * sized by the caller (gate tests use >> 256 workgroups to fill the 8 XCDs).
 */
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#define CHECK(x)                                                   \
  do {                                                             \
    hipError_t _e = (x);                                           \
    if (_e != hipSuccess) {                                        \
      fprintf(stderr, "probe: %s failed: %d (%s)\n", #x, (int)_e,  \
              hipGetErrorString(_e));                              \
      return (int)_e;                                              \
    }                                                              \
  } while (0)

__global__ void vecadd_kernel(const float *a, const float *b, float *c,
                              size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) c[i] = a[i] + b[i];
}

__global__ void burn_kernel(uint64_t cycles, float *sink) {
  uint64_t start = clock64();
  float acc = threadIdx.x * 1e-6f;
  while (clock64() - start < cycles) {
    /* keep the VALU busy so gpu_busy_percent reflects real occupancy */
    for (int k = 0; k < 256; k++) acc = fmaf(acc, 1.000001f, 1e-7f);
  }
  if (acc == 12345.678f) *sink = acc; /* never true; defeats DCE */
}

__global__ void tiny_kernel(float *sink) {
  if (threadIdx.x == 0 && blockIdx.x == 0) sink[0] += 1.0f;
}

extern "C" {

int vgpu_probe_vecadd(size_t n) {
  float *a, *b, *c;
  CHECK(hipMalloc((void **)&a, n * sizeof(float)));
  CHECK(hipMalloc((void **)&b, n * sizeof(float)));
  CHECK(hipMalloc((void **)&c, n * sizeof(float)));
  float *ha = new float[n], *hb = new float[n], *hc = new float[n];
  for (size_t i = 0; i < n; i++) {
    ha[i] = (float)(i % 977);
    hb[i] = 1.5f;
  }
  CHECK(hipMemcpy(a, ha, n * sizeof(float), hipMemcpyHostToDevice));
  CHECK(hipMemcpy(b, hb, n * sizeof(float), hipMemcpyHostToDevice));
  dim3 block(256), grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(vecadd_kernel, grid, block, 0, 0, a, b, c, n);
  CHECK(hipGetLastError());
  CHECK(hipMemcpy(hc, c, n * sizeof(float), hipMemcpyDeviceToHost));
  int bad = 0;
  for (size_t i = 0; i < n; i++)
    if (hc[i] != ha[i] + 1.5f) bad++;
  delete[] ha;
  delete[] hb;
  delete[] hc;
  CHECK(hipFree(a));
  CHECK(hipFree(b));
  CHECK(hipFree(c));
  return bad ? -1 : 0;
}

/* Launch `iters` burn kernels of `grid` workgroups, each spinning for
 * about `ms_per_kernel` milliseconds at 2.4 GHz.  Returns wall seconds. */
double vgpu_probe_burn(int iters, int grid, int ms_per_kernel) {
  float *sink;
  if (hipMalloc((void **)&sink, sizeof(float)) != hipSuccess) return -1.0;
  uint64_t cycles = (uint64_t)ms_per_kernel * 2400000ULL;
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, 0);
  for (int i = 0; i < iters; i++)
    hipLaunchKernelGGL(burn_kernel, dim3(grid), dim3(256), 0, 0, cycles, sink);
  (void)hipEventRecord(t1, 0);
  if (hipEventSynchronize(t1) != hipSuccess) return -1.0;
  float ms = 0;
  (void)hipEventElapsedTime(&ms, t0, t1);
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  (void)hipFree(sink);
  return ms / 1000.0;
}

/* n tiny launches of `grid` workgroups; returns wall seconds (pacing test) */
double vgpu_probe_storm(long n, int grid) {
  float *sink;
  if (hipMalloc((void **)&sink, sizeof(float)) != hipSuccess) return -1.0;
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, 0);
  for (long i = 0; i < n; i++)
    hipLaunchKernelGGL(tiny_kernel, dim3(grid), dim3(64), 0, 0, sink);
  (void)hipEventRecord(t1, 0);
  if (hipEventSynchronize(t1) != hipSuccess) return -1.0;
  float ms = 0;
  (void)hipEventElapsedTime(&ms, t0, t1);
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  (void)hipFree(sink);
  return ms / 1000.0;
}

int vgpu_probe_device_count(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return -1;
  return n;
}

} /* extern "C" */
