// bw_sweep.hip — HBM streaming-copy parameter sweep on MI355X (gfx950).
//
// Finds the best (block, grid, access pattern) for the GPU validator's
// bandwidth check.  Reference point: MI355X_MICROARCH.md reports 6.29 TB/s
// measured float4 copy (79% of the 8 TB/s spec).
//
// Build: hipcc --offload-arch=gfx950 -O3 experiments/bw_sweep.hip -o gpurun_out/bw_sweep
// Run:   ./gpurun_out/bw_sweep [buf_mib]

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define HIP_CHECK(x)                                                    \
  do {                                                                  \
    hipError_t e = (x);                                                 \
    if (e != hipSuccess) {                                              \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e),  \
              __FILE__, __LINE__);                                      \
      exit(1);                                                          \
    }                                                                   \
  } while (0)

// grid-stride float4 (the current validator kernel)
__global__ void copy_gridstride(const float4* __restrict__ src,
                                float4* __restrict__ dst, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = src[i];
}

// grid-stride with 2x unroll (independent loads in flight)
__global__ void copy_unroll2(const float4* __restrict__ src,
                             float4* __restrict__ dst, size_t n) {
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + stride < n; i += 2 * stride) {
    float4 a = src[i];
    float4 b = src[i + stride];
    dst[i] = a;
    dst[i + stride] = b;
  }
  for (; i < n; i += stride) dst[i] = src[i];
}

// grid-stride with 4x unroll
__global__ void copy_unroll4(const float4* __restrict__ src,
                             float4* __restrict__ dst, size_t n) {
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + 3 * stride < n; i += 4 * stride) {
    float4 a = src[i];
    float4 b = src[i + stride];
    float4 c = src[i + 2 * stride];
    float4 d = src[i + 3 * stride];
    dst[i] = a;
    dst[i + stride] = b;
    dst[i + 2 * stride] = c;
    dst[i + 3 * stride] = d;
  }
  for (; i < n; i += stride) dst[i] = src[i];
}

// nontemporal load/store (bypass-cache streaming hints) with 2x unroll.
// The builtin needs a true vector type, not HIP_vector_type.
typedef float vfloat4 __attribute__((ext_vector_type(4)));

__global__ void copy_nt2(const float4* __restrict__ src4,
                         float4* __restrict__ dst4, size_t n) {
  const vfloat4* __restrict__ src = reinterpret_cast<const vfloat4*>(src4);
  vfloat4* __restrict__ dst = reinterpret_cast<vfloat4*>(dst4);
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + stride < n; i += 2 * stride) {
    vfloat4 a = __builtin_nontemporal_load(&src[i]);
    vfloat4 b = __builtin_nontemporal_load(&src[i + stride]);
    __builtin_nontemporal_store(a, &dst[i]);
    __builtin_nontemporal_store(b, &dst[i + stride]);
  }
  for (; i < n; i += stride) dst[i] = src[i];
}

using Kernel = void (*)(const float4*, float4*, size_t);

static double run_case(Kernel k, const float4* src, float4* dst, size_t n,
                       int grid, int block, int iters) {
  hipLaunchKernelGGL(k, dim3(grid), dim3(block), 0, 0, src, dst, n);  // warmup
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL(k, dim3(grid), dim3(block), 0, 0, src, dst, n);
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  double bytes = 2.0 * n * sizeof(float4) * iters;
  return bytes / 1e9 / (ms / 1e3);
}

int main(int argc, char** argv) {
  double buf_mib = argc > 1 ? atof(argv[1]) : 2048.0;
  size_t n = (size_t)(buf_mib * 1024 * 1024) / sizeof(float4);
  float4 *src, *dst;
  HIP_CHECK(hipMalloc(&src, n * sizeof(float4)));
  HIP_CHECK(hipMalloc(&dst, n * sizeof(float4)));
  HIP_CHECK(hipMemset(src, 1, n * sizeof(float4)));

  struct { const char* name; Kernel k; } kernels[] = {
      {"gridstride", copy_gridstride},
      {"unroll2", copy_unroll2},
      {"nt2", copy_nt2},
  };
  int blocks[] = {512, 1024};
  int grids[] = {16384, 32768, 65536, 131072};
  printf("buf=%.0f MiB (%zu float4)\n", buf_mib, n);
  printf("%-12s %6s %7s %10s\n", "kernel", "block", "grid", "GB/s");
  double best = 0;
  const char* best_desc = "";
  static char desc[128];
  for (auto& kk : kernels) {
    for (int b : blocks) {
      for (int g : grids) {
        double gbps = run_case(kk.k, src, dst, n, g, b, 5);
        printf("%-12s %6d %7d %10.0f\n", kk.name, b, g, gbps);
        if (gbps > best) {
          best = gbps;
          snprintf(desc, sizeof desc, "%s block=%d grid=%d", kk.name, b, g);
          best_desc = desc;
        }
      }
    }
  }
  printf("BEST: %s -> %.0f GB/s\n", best_desc, best);
  HIP_CHECK(hipFree(src));
  HIP_CHECK(hipFree(dst));
  return 0;
}
