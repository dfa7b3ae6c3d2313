// CDNA4 (gfx950) local-reduction kernels for the ring/tree collectives.
//
// Replaces the reference's THC axpy-style local add inside its p2p allreduce
// (TorchMPI lib/collectives_cuda.cpp chunk `dst += src`, SURVEY.md section
// 2.4 row 7). Design for MI355X:
//   - 64-wide wavefronts, 256-thread blocks, grid capped at 2048 blocks with
//     a grid-stride loop (memory-bound op; guide section 6 Guideline 11).
//   - 16 B/lane vectorized loads/stores (dwordx4) — hipcc does not
//     auto-vectorize bf16 loads (Guideline 13).
//   - bf16/f16 accumulate in fp32 in-register, round-to-nearest-even on
//     store (documented error bound: one rounding per ring step).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#include "../common.h"

namespace tma {

namespace {

constexpr int kBlock = 256;
constexpr int kMaxBlocks = 2048;  // 256 CU x 8 blocks/CU

template <typename T>
struct AccOf {
  using type = T;
};
template <>
struct AccOf<__hip_bfloat16> {
  using type = float;
};
template <>
struct AccOf<__half> {
  using type = float;
};

template <typename T>
__device__ __forceinline__ typename AccOf<T>::type to_acc(T v) {
  return static_cast<typename AccOf<T>::type>(v);
}
template <>
__device__ __forceinline__ float to_acc<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <>
__device__ __forceinline__ float to_acc<__half>(__half v) {
  return __half2float(v);
}

template <typename T>
__device__ __forceinline__ T from_acc(typename AccOf<T>::type v) {
  return static_cast<T>(v);
}
template <>
__device__ __forceinline__ __hip_bfloat16 from_acc<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}
template <>
__device__ __forceinline__ __half from_acc<__half>(float v) {
  return __float2half(v);
}

template <typename A>
__device__ __forceinline__ A combine(A a, A b, int op) {
  switch (op) {
    case 0: return a + b;
    case 1: return a > b ? a : b;
    case 2: return a < b ? a : b;
    default: return a * b;
  }
}

// dst[i] = dst[i] (op) src[i], 16-byte vectorized.
template <typename T, int OP>
__global__ __launch_bounds__(kBlock) void reduce_vec_kernel(T* __restrict__ dst,
                                                            const T* __restrict__ src,
                                                            size_t n) {
  constexpr int VEC = 16 / sizeof(T);
  using V = __attribute__((ext_vector_type(VEC * sizeof(T) / 4))) unsigned int;
  const size_t nvec = n / VEC;
  const size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  size_t i = static_cast<size_t>(blockIdx.x) * blockDim.x + threadIdx.x;

  for (size_t v = i; v < nvec; v += stride) {
    V dv = reinterpret_cast<V*>(dst)[v];
    V sv = reinterpret_cast<const V*>(src)[v];
    T* dp = reinterpret_cast<T*>(&dv);
    const T* sp = reinterpret_cast<const T*>(&sv);
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      dp[k] = from_acc<T>(combine(to_acc(dp[k]), to_acc(sp[k]), OP));
    }
    reinterpret_cast<V*>(dst)[v] = dv;
  }
  // Scalar tail.
  size_t tail = nvec * VEC + i;
  if (tail < n) {
    dst[tail] = from_acc<T>(combine(to_acc(dst[tail]), to_acc(src[tail]), OP));
  }
}

// Unaligned fallback: scalar grid-stride.
template <typename T, int OP>
__global__ __launch_bounds__(kBlock) void reduce_scalar_kernel(T* __restrict__ dst,
                                                               const T* __restrict__ src,
                                                               size_t n) {
  const size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (size_t i = static_cast<size_t>(blockIdx.x) * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    dst[i] = from_acc<T>(combine(to_acc(dst[i]), to_acc(src[i]), OP));
  }
}

// data[i] *= alpha (float types only; used for the 1/W average epilogue).
template <typename T>
__global__ __launch_bounds__(kBlock) void scale_vec_kernel(T* __restrict__ data,
                                                           size_t n, float alpha) {
  constexpr int VEC = 16 / sizeof(T);
  using V = __attribute__((ext_vector_type(VEC * sizeof(T) / 4))) unsigned int;
  const size_t nvec = n / VEC;
  const size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  size_t i = static_cast<size_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  for (size_t v = i; v < nvec; v += stride) {
    V dv = reinterpret_cast<V*>(data)[v];
    T* dp = reinterpret_cast<T*>(&dv);
#pragma unroll
    for (int k = 0; k < VEC; ++k)
      dp[k] = from_acc<T>(to_acc(dp[k]) * static_cast<typename AccOf<T>::type>(alpha));
    reinterpret_cast<V*>(data)[v] = dv;
  }
  size_t tail = nvec * VEC + i;
  if (tail < n)
    data[tail] =
        from_acc<T>(to_acc(data[tail]) * static_cast<typename AccOf<T>::type>(alpha));
}

// f64 keeps alpha in double; other types take float (AccOf is float/f32).
__global__ __launch_bounds__(kBlock) void scale_vec_f64_kernel(double* __restrict__ data,
                                                               size_t n, double alpha) {
  const size_t nvec = n / 2;
  const size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  size_t i = static_cast<size_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  for (size_t v = i; v < nvec; v += stride) {
    double2 dv = reinterpret_cast<double2*>(data)[v];
    dv.x *= alpha;
    dv.y *= alpha;
    reinterpret_cast<double2*>(data)[v] = dv;
  }
  size_t tail = nvec * 2 + i;
  if (tail < n) data[tail] *= alpha;
}

// Unaligned fallback: scalar grid-stride (mirrors launch_reduce's dispatch).
template <typename T, typename A>
__global__ __launch_bounds__(kBlock) void scale_scalar_kernel(T* __restrict__ data,
                                                              size_t n, A alpha) {
  const size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (size_t i = static_cast<size_t>(blockIdx.x) * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    data[i] = from_acc<T>(to_acc(data[i]) * static_cast<typename AccOf<T>::type>(alpha));
  }
}

// Measured on MI355X (benchmarks/probe_reduce.hip, profiles/): at >= 64 MB
// streamed, 512 blocks (2/CU) beats 2048 by ~8% (less DVFS/L2 pressure);
// below that the larger grid wins slightly.
inline int grid_for(size_t work_items, size_t bytes_streamed) {
  size_t cap = bytes_streamed >= (64u << 20) ? 512 : kMaxBlocks;
  size_t blocks = (work_items + kBlock - 1) / kBlock;
  if (blocks > cap) blocks = cap;
  if (blocks == 0) blocks = 1;
  return static_cast<int>(blocks);
}

template <typename T>
void launch_reduce_t(void* dst, const void* src, size_t n, RedOp op,
                     hipStream_t stream) {
  bool aligned = ((reinterpret_cast<uintptr_t>(dst) | reinterpret_cast<uintptr_t>(src)) &
                  15) == 0;
  constexpr int VEC = 16 / sizeof(T);
  int grid = grid_for(aligned ? (n / VEC + 1) : n, 3 * n * sizeof(T));
  auto d = static_cast<T*>(dst);
  auto s = static_cast<const T*>(src);
#define TMA_DISPATCH_OP(OPI)                                                          \
  if (aligned)                                                                        \
    hipLaunchKernelGGL((reduce_vec_kernel<T, OPI>), dim3(grid), dim3(kBlock), 0,      \
                       stream, d, s, n);                                              \
  else                                                                                \
    hipLaunchKernelGGL((reduce_scalar_kernel<T, OPI>), dim3(grid), dim3(kBlock), 0,   \
                       stream, d, s, n)
  switch (op) {
    case RedOp::SUM: TMA_DISPATCH_OP(0); break;
    case RedOp::MAX: TMA_DISPATCH_OP(1); break;
    case RedOp::MIN: TMA_DISPATCH_OP(2); break;
    case RedOp::PROD: TMA_DISPATCH_OP(3); break;
  }
#undef TMA_DISPATCH_OP
}

}  // namespace

// Public launchers -----------------------------------------------------------

void launch_reduce(void* dst, const void* src, size_t count, DType d, RedOp op,
                   hipStream_t stream) {
  switch (d) {
    case DType::F32: launch_reduce_t<float>(dst, src, count, op, stream); break;
    case DType::F64: launch_reduce_t<double>(dst, src, count, op, stream); break;
    case DType::BF16: launch_reduce_t<__hip_bfloat16>(dst, src, count, op, stream); break;
    case DType::F16: launch_reduce_t<__half>(dst, src, count, op, stream); break;
    case DType::I32: launch_reduce_t<int32_t>(dst, src, count, op, stream); break;
    case DType::I64: launch_reduce_t<int64_t>(dst, src, count, op, stream); break;
    case DType::U8: launch_reduce_t<uint8_t>(dst, src, count, op, stream); break;
    case DType::I8: launch_reduce_t<int8_t>(dst, src, count, op, stream); break;
  }
}

void launch_scale(void* data, size_t count, DType d, double alpha, hipStream_t stream) {
  int grid = grid_for(count / 4 + 1, 2 * count * dtype_size(d));
  // A storage-offset view can be contiguous but not 16-byte aligned; the
  // vector kernels require 16 B (same dispatch rule as launch_reduce).
  bool aligned = (reinterpret_cast<uintptr_t>(data) & 15) == 0;
  switch (d) {
    case DType::F32:
      if (aligned)
        hipLaunchKernelGGL((scale_vec_kernel<float>), dim3(grid), dim3(kBlock), 0,
                           stream, static_cast<float*>(data), count,
                           static_cast<float>(alpha));
      else
        hipLaunchKernelGGL((scale_scalar_kernel<float, float>), dim3(grid),
                           dim3(kBlock), 0, stream, static_cast<float*>(data), count,
                           static_cast<float>(alpha));
      break;
    case DType::F64:
      if (aligned)
        hipLaunchKernelGGL(scale_vec_f64_kernel, dim3(grid), dim3(kBlock), 0, stream,
                           static_cast<double*>(data), count, alpha);
      else
        hipLaunchKernelGGL((scale_scalar_kernel<double, double>), dim3(grid),
                           dim3(kBlock), 0, stream, static_cast<double*>(data), count,
                           alpha);
      break;
    case DType::BF16:
      if (aligned)
        hipLaunchKernelGGL((scale_vec_kernel<__hip_bfloat16>), dim3(grid), dim3(kBlock),
                           0, stream, static_cast<__hip_bfloat16*>(data), count,
                           static_cast<float>(alpha));
      else
        hipLaunchKernelGGL((scale_scalar_kernel<__hip_bfloat16, float>), dim3(grid),
                           dim3(kBlock), 0, stream, static_cast<__hip_bfloat16*>(data),
                           count, static_cast<float>(alpha));
      break;
    case DType::F16:
      if (aligned)
        hipLaunchKernelGGL((scale_vec_kernel<__half>), dim3(grid), dim3(kBlock), 0,
                           stream, static_cast<__half*>(data), count,
                           static_cast<float>(alpha));
      else
        hipLaunchKernelGGL((scale_scalar_kernel<__half, float>), dim3(grid),
                           dim3(kBlock), 0, stream, static_cast<__half*>(data), count,
                           static_cast<float>(alpha));
      break;
    default:
      throw std::runtime_error("torchmpi_amd: scale supports float dtypes only");
  }
}

}  // namespace tma
