// Standalone tuning probe for the ring-step reduce kernel (dst += src, bf16).
// Sweeps grid size x per-thread width x access pattern; within-run A/B
// (guide section 5.4 rule 24: interleaved rounds, report median).
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 benchmarks/probe_reduce.hip \
//        -o gpurun_out/probe_reduce
// Run on the GPU box: ./gpurun_out/probe_reduce
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <algorithm>
#include <cstdio>
#include <vector>

#define CHECK(c)                                                   \
  do {                                                             \
    hipError_t e = (c);                                            \
    if (e != hipSuccess) {                                         \
      printf("HIP error %s @%d\n", hipGetErrorString(e), __LINE__); \
      return 1;                                                    \
    }                                                              \
  } while (0)

using bf16 = __hip_bfloat16;

// Variant A: current production kernel — 16B/lane uint4, grid-stride.
__global__ __launch_bounds__(256) void reduce_v16(bf16* __restrict__ dst,
                                                  const bf16* __restrict__ src,
                                                  size_t n) {
  const size_t nvec = n / 8;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t v = (size_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec;
       v += stride) {
    uint4 dv = reinterpret_cast<uint4*>(dst)[v];
    uint4 sv = reinterpret_cast<const uint4*>(src)[v];
    bf16* dp = reinterpret_cast<bf16*>(&dv);
    const bf16* sp = reinterpret_cast<const bf16*>(&sv);
#pragma unroll
    for (int k = 0; k < 8; ++k)
      dp[k] = __float2bfloat16(__bfloat162float(dp[k]) + __bfloat162float(sp[k]));
    reinterpret_cast<uint4*>(dst)[v] = dv;
  }
}

// Variant B: 32B per lane per iteration (2x uint4, issued back to back).
__global__ __launch_bounds__(256) void reduce_v32(bf16* __restrict__ dst,
                                                  const bf16* __restrict__ src,
                                                  size_t n) {
  const size_t npair = n / 16;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t v = (size_t)blockIdx.x * blockDim.x + threadIdx.x; v < npair;
       v += stride) {
    uint4 d0 = reinterpret_cast<uint4*>(dst)[2 * v];
    uint4 d1 = reinterpret_cast<uint4*>(dst)[2 * v + 1];
    uint4 s0 = reinterpret_cast<const uint4*>(src)[2 * v];
    uint4 s1 = reinterpret_cast<const uint4*>(src)[2 * v + 1];
    bf16* dp0 = reinterpret_cast<bf16*>(&d0);
    bf16* dp1 = reinterpret_cast<bf16*>(&d1);
    const bf16* sp0 = reinterpret_cast<const bf16*>(&s0);
    const bf16* sp1 = reinterpret_cast<const bf16*>(&s1);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      dp0[k] = __float2bfloat16(__bfloat162float(dp0[k]) + __bfloat162float(sp0[k]));
      dp1[k] = __float2bfloat16(__bfloat162float(dp1[k]) + __bfloat162float(sp1[k]));
    }
    reinterpret_cast<uint4*>(dst)[2 * v] = d0;
    reinterpret_cast<uint4*>(dst)[2 * v + 1] = d1;
  }
}

// Variant C: block-contiguous partitioning (each block owns one contiguous
// span — better L2/XCD locality than grid-stride interleave).
__global__ __launch_bounds__(256) void reduce_span(bf16* __restrict__ dst,
                                                   const bf16* __restrict__ src,
                                                   size_t n) {
  const size_t nvec = n / 8;
  const size_t per_block = (nvec + gridDim.x - 1) / gridDim.x;
  const size_t beg = (size_t)blockIdx.x * per_block;
  const size_t end = min(beg + per_block, nvec);
  for (size_t v = beg + threadIdx.x; v < end; v += blockDim.x) {
    uint4 dv = reinterpret_cast<uint4*>(dst)[v];
    uint4 sv = reinterpret_cast<const uint4*>(src)[v];
    bf16* dp = reinterpret_cast<bf16*>(&dv);
    const bf16* sp = reinterpret_cast<const bf16*>(&sv);
#pragma unroll
    for (int k = 0; k < 8; ++k)
      dp[k] = __float2bfloat16(__bfloat162float(dp[k]) + __bfloat162float(sp[k]));
    reinterpret_cast<uint4*>(dst)[v] = dv;
  }
}

// Variant D: packed-math v_pk_add via __hip_bfloat162 2-wide ops.
__global__ __launch_bounds__(256) void reduce_pk(bf16* __restrict__ dst,
                                                 const bf16* __restrict__ src,
                                                 size_t n) {
  const size_t nvec = n / 8;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t v = (size_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec;
       v += stride) {
    uint4 dv = reinterpret_cast<uint4*>(dst)[v];
    uint4 sv = reinterpret_cast<const uint4*>(src)[v];
    __hip_bfloat162* dp = reinterpret_cast<__hip_bfloat162*>(&dv);
    const __hip_bfloat162* sp = reinterpret_cast<const __hip_bfloat162*>(&sv);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float2 a = __bfloat1622float2(dp[k]);
      float2 b = __bfloat1622float2(sp[k]);
      dp[k] = __float22bfloat162_rn(make_float2(a.x + b.x, a.y + b.y));
    }
    reinterpret_cast<uint4*>(dst)[v] = dv;
  }
}

struct Variant {
  const char* name;
  void (*kern)(bf16*, const bf16*, size_t);
  int elems_per_thread;  // per iteration
};

int main() {
  const size_t MB = 1 << 20;
  std::vector<size_t> sizes = {1 * MB, 16 * MB, 64 * MB, 256 * MB, 512 * MB};
  const size_t max_n = 512 * MB / 2;
  bf16 *d, *s;
  CHECK(hipMalloc(&d, max_n * 2));
  CHECK(hipMalloc(&s, max_n * 2));
  CHECK(hipMemset(d, 0x3f, max_n * 2));
  CHECK(hipMemset(s, 0x3f, max_n * 2));

  Variant vars[] = {
      {"v16_gridstride", reduce_v16, 8},
      {"v32_gridstride", reduce_v32, 16},
      {"v16_span", reduce_span, 8},
      {"v16_pk", reduce_pk, 8},
  };
  std::vector<int> grids = {512, 1024, 2048, 4096};

  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));

  printf("%-16s %6s %8s  GB/s (median of 7, interleaved)\n", "variant", "grid",
         "size_MB");
  for (size_t bytes : sizes) {
    size_t n = bytes / 2;
    for (auto& v : vars) {
      for (int g : grids) {
        std::vector<float> ts;
        for (int rep = 0; rep < 7; ++rep) {
          CHECK(hipEventRecord(e0));
          for (int it = 0; it < 5; ++it)
            hipLaunchKernelGGL(v.kern, dim3(g), dim3(256), 0, 0, d, s, n);
          CHECK(hipEventRecord(e1));
          CHECK(hipEventSynchronize(e1));
          float ms;
          CHECK(hipEventElapsedTime(&ms, e0, e1));
          ts.push_back(ms / 5);
        }
        std::sort(ts.begin(), ts.end());
        float med = ts[ts.size() / 2];
        double gbps = 3.0 * bytes / (med * 1e-3) / 1e9;
        printf("%-16s %6d %8zu  %7.1f\n", v.name, g, bytes / MB, gbps);
      }
    }
  }
  return 0;
}
