// Fused NHWC (channels_last) BatchNorm for gfx950 — training fwd + bwd.
//
// Motivation (profiles/resnet50_bf16_kernel_stats.md): MIOpen's spatial
// BatchNorm kernels are ~35% of the ResNet-50 bf16 step on MI355X. These
// kernels exploit the channels-last layout directly:
//   - stats/reduction passes: lanes own CONSECUTIVE CHANNELS (64 channels =
//     one wave = one coalesced 128 B bf16 load per row), fp32 accumulation,
//     one global atomicAdd per (block x channel),
//   - apply passes: 16 B/lane vectorized over channels (C % 8 == 0 for
//     bf16; scalar fallback otherwise).
// Stats and affine params are fp32; x/y/dy may be bf16 or f32.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <ATen/ATen.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <torch/extension.h>

#include "../common.h"

namespace tma {

#ifndef TMA_HIP_CHECK
#define TMA_HIP_CHECK(cmd)                                                    \
  do {                                                                        \
    hipError_t e = (cmd);                                                     \
    TMA_CHECK(e == hipSuccess, std::string("HIP error: ") + hipGetErrorString(e)); \
  } while (0)
#endif

namespace {

constexpr int kLanes = 64;   // channels per wave
constexpr int kRowsPB = 4;   // rows (N*H*W positions) per block iteration

template <typename T>
__device__ __forceinline__ float ldf(T v);
template <>
__device__ __forceinline__ float ldf<float>(float v) { return v; }
template <>
__device__ __forceinline__ float ldf<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}

template <typename T>
__device__ __forceinline__ T stf(float v);
template <>
__device__ __forceinline__ float stf<float>(float v) { return v; }
template <>
__device__ __forceinline__ __hip_bfloat16 stf<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// ---- forward stats: per-channel sum / sumsq ------------------------------
//
// Vectorized form (VEC = 8 bf16 / 4 f32 channels = 16 B per lane): a block of
// 256 threads is laid out as (c8_per_block channel-groups x rows); because
// c8_per_block divides 256 and nC8 * 16 B = one row's bytes, a wave's 64
// lanes cover contiguous memory for ANY power-of-two C — full-rate loads.
// Per-thread fp32 accumulators (VEC of them), block LDS reduction over the
// row copies, one atomicAdd per (block x channel).

template <typename T, int VEC>
__global__ __launch_bounds__(256) void bn_stats_vec_kernel(
    const T* __restrict__ x, float* __restrict__ sum, float* __restrict__ sumsq,
    size_t M, int nC8, int c8_per_block, int row_blocks) {
  const int c_blocks = nC8 / c8_per_block;
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int rows_per_iter = 256 / c8_per_block;
  const int c8l = threadIdx.x % c8_per_block;          // channel-group in block
  const int rsub = threadIdx.x / c8_per_block;         // row copy
  const int c8 = cb * c8_per_block + c8l;
  float acc[VEC], accsq[VEC];
#pragma unroll
  for (int e = 0; e < VEC; ++e) acc[e] = accsq[e] = 0.f;
  const size_t row_stride = (size_t)row_blocks * rows_per_iter;
  const T* base = x + (size_t)c8 * VEC;
  for (size_t r = (size_t)rb * rows_per_iter + rsub; r < M; r += row_stride) {
    T v[VEC];
    *reinterpret_cast<uint4*>(v) =
        *reinterpret_cast<const uint4*>(base + r * (size_t)nC8 * VEC);
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      float f = ldf(v[e]);
      acc[e] += f;
      accsq[e] += f * f;
    }
  }
  // LDS reduction across the row copies of each channel-group.
  __shared__ float ls[256 * VEC];
#pragma unroll
  for (int e = 0; e < VEC; ++e) ls[threadIdx.x * VEC + e] = acc[e];
  __syncthreads();
  if (rsub == 0) {
    for (int w = 1; w < rows_per_iter; ++w)
#pragma unroll
      for (int e = 0; e < VEC; ++e)
        acc[e] += ls[(w * c8_per_block + c8l) * VEC + e];
  }
  __syncthreads();
#pragma unroll
  for (int e = 0; e < VEC; ++e) ls[threadIdx.x * VEC + e] = accsq[e];
  __syncthreads();
  if (rsub == 0) {
    for (int w = 1; w < rows_per_iter; ++w)
#pragma unroll
      for (int e = 0; e < VEC; ++e)
        accsq[e] += ls[(w * c8_per_block + c8l) * VEC + e];
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      atomicAdd(&sum[c8 * VEC + e], acc[e]);
      atomicAdd(&sumsq[c8 * VEC + e], accsq[e]);
    }
  }
}

// Scalar fallback for C not divisible by the vector width / 256 layout.
template <typename T>
__global__ __launch_bounds__(kLanes* kRowsPB) void bn_stats_kernel(
    const T* __restrict__ x, float* __restrict__ sum, float* __restrict__ sumsq,
    size_t M, int C, int c_blocks, int row_blocks) {
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int c = cb * kLanes + threadIdx.x;
  float s = 0.f, ss = 0.f;
  if (c < C) {
    const size_t row_stride = (size_t)row_blocks * kRowsPB;
    for (size_t r = (size_t)rb * kRowsPB + threadIdx.y; r < M; r += row_stride) {
      float v = ldf(x[r * C + c]);
      s += v;
      ss += v * v;
    }
  }
  // reduce the kRowsPB partials per channel through LDS
  __shared__ float ls[kRowsPB][kLanes];
  __shared__ float lss[kRowsPB][kLanes];
  ls[threadIdx.y][threadIdx.x] = s;
  lss[threadIdx.y][threadIdx.x] = ss;
  __syncthreads();
  if (threadIdx.y == 0 && c < C) {
    float ts = 0.f, tss = 0.f;
#pragma unroll
    for (int w = 0; w < kRowsPB; ++w) {
      ts += ls[w][threadIdx.x];
      tss += lss[w][threadIdx.x];
    }
    atomicAdd(&sum[c], ts);
    atomicAdd(&sumsq[c], tss);
  }
}

// ---- finalize: mean/invstd + running stats -------------------------------

__global__ void bn_finalize_kernel(const float* __restrict__ sum,
                                   const float* __restrict__ sumsq,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var, float momentum,
                                   float eps, float M, int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float m = sum[c] / M;
  float var = fmaxf(sumsq[c] / M - m * m, 0.f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    float unbiased = M > 1.f ? var * M / (M - 1.f) : var;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// ---- forward apply -------------------------------------------------------

template <typename T, int VEC>
__global__ __launch_bounds__(256) void bn_fwd_apply_kernel(
    const T* __restrict__ x, T* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ weight,
    const float* __restrict__ bias, size_t total, int C) {
  const size_t nvec = total / VEC;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t v = (size_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec;
       v += stride) {
    size_t idx = v * VEC;
    int c0 = (int)(idx % C);
    T in[VEC], out[VEC];
    *reinterpret_cast<uint4*>(in) = reinterpret_cast<const uint4*>(x)[v];
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      int c = c0 + e;
      float xh = (ldf(in[e]) - mean[c]) * invstd[c];
      out[e] = stf<T>(xh * weight[c] + bias[c]);
    }
    reinterpret_cast<uint4*>(y)[v] = *reinterpret_cast<uint4*>(out);
  }
}

template <typename T>
__global__ __launch_bounds__(256) void bn_fwd_apply_scalar(
    const T* __restrict__ x, T* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ weight,
    const float* __restrict__ bias, size_t total, int C) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    float xh = (ldf(x[i]) - mean[c]) * invstd[c];
    y[i] = stf<T>(xh * weight[c] + bias[c]);
  }
}

// ---- backward stats: sum(dy), sum(dy * xhat) -----------------------------

template <typename T, int VEC>
__global__ __launch_bounds__(256) void bn_bwd_stats_vec_kernel(
    const T* __restrict__ x, const T* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    float* __restrict__ sdy, float* __restrict__ sdyx, size_t M, int nC8,
    int c8_per_block, int row_blocks) {
  const int c_blocks = nC8 / c8_per_block;
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int rows_per_iter = 256 / c8_per_block;
  const int c8l = threadIdx.x % c8_per_block;
  const int rsub = threadIdx.x / c8_per_block;
  const int c8 = cb * c8_per_block + c8l;
  float mu[VEC], is[VEC], acc[VEC], accx[VEC];
#pragma unroll
  for (int e = 0; e < VEC; ++e) {
    mu[e] = mean[c8 * VEC + e];
    is[e] = invstd[c8 * VEC + e];
    acc[e] = accx[e] = 0.f;
  }
  const size_t row_stride = (size_t)row_blocks * rows_per_iter;
  const T* xb = x + (size_t)c8 * VEC;
  const T* gb = dy + (size_t)c8 * VEC;
  for (size_t r = (size_t)rb * rows_per_iter + rsub; r < M; r += row_stride) {
    T xv[VEC], gv[VEC];
    *reinterpret_cast<uint4*>(xv) =
        *reinterpret_cast<const uint4*>(xb + r * (size_t)nC8 * VEC);
    *reinterpret_cast<uint4*>(gv) =
        *reinterpret_cast<const uint4*>(gb + r * (size_t)nC8 * VEC);
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      float g = ldf(gv[e]);
      float xh = (ldf(xv[e]) - mu[e]) * is[e];
      acc[e] += g;
      accx[e] += g * xh;
    }
  }
  __shared__ float ls[256 * VEC];
#pragma unroll
  for (int e = 0; e < VEC; ++e) ls[threadIdx.x * VEC + e] = acc[e];
  __syncthreads();
  if (rsub == 0) {
    for (int w = 1; w < rows_per_iter; ++w)
#pragma unroll
      for (int e = 0; e < VEC; ++e)
        acc[e] += ls[(w * c8_per_block + c8l) * VEC + e];
  }
  __syncthreads();
#pragma unroll
  for (int e = 0; e < VEC; ++e) ls[threadIdx.x * VEC + e] = accx[e];
  __syncthreads();
  if (rsub == 0) {
    for (int w = 1; w < rows_per_iter; ++w)
#pragma unroll
      for (int e = 0; e < VEC; ++e)
        accx[e] += ls[(w * c8_per_block + c8l) * VEC + e];
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      atomicAdd(&sdy[c8 * VEC + e], acc[e]);
      atomicAdd(&sdyx[c8 * VEC + e], accx[e]);
    }
  }
}

template <typename T>
__global__ __launch_bounds__(kLanes* kRowsPB) void bn_bwd_stats_kernel(
    const T* __restrict__ x, const T* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    float* __restrict__ sdy, float* __restrict__ sdyx, size_t M, int C,
    int c_blocks, int row_blocks) {
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int c = cb * kLanes + threadIdx.x;
  float s = 0.f, sx = 0.f;
  if (c < C) {
    const float mu = mean[c], is = invstd[c];
    const size_t row_stride = (size_t)row_blocks * kRowsPB;
    for (size_t r = (size_t)rb * kRowsPB + threadIdx.y; r < M; r += row_stride) {
      float g = ldf(dy[r * C + c]);
      float xh = (ldf(x[r * C + c]) - mu) * is;
      s += g;
      sx += g * xh;
    }
  }
  __shared__ float ls[kRowsPB][kLanes];
  __shared__ float lsx[kRowsPB][kLanes];
  ls[threadIdx.y][threadIdx.x] = s;
  lsx[threadIdx.y][threadIdx.x] = sx;
  __syncthreads();
  if (threadIdx.y == 0 && c < C) {
    float ts = 0.f, tsx = 0.f;
#pragma unroll
    for (int w = 0; w < kRowsPB; ++w) {
      ts += ls[w][threadIdx.x];
      tsx += lsx[w][threadIdx.x];
    }
    atomicAdd(&sdy[c], ts);
    atomicAdd(&sdyx[c], tsx);
  }
}

// ---- backward apply ------------------------------------------------------

template <typename T, int VEC>
__global__ __launch_bounds__(256) void bn_bwd_apply_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, T* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, const float* __restrict__ sdy,
    const float* __restrict__ sdyx, float invM, size_t total, int C) {
  const size_t nvec = total / VEC;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t v = (size_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec;
       v += stride) {
    size_t idx = v * VEC;
    int c0 = (int)(idx % C);
    T xin[VEC], gin[VEC], out[VEC];
    *reinterpret_cast<uint4*>(xin) = reinterpret_cast<const uint4*>(x)[v];
    *reinterpret_cast<uint4*>(gin) = reinterpret_cast<const uint4*>(dy)[v];
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      int c = c0 + e;
      float xh = (ldf(xin[e]) - mean[c]) * invstd[c];
      float g = ldf(gin[e]);
      float t = g - sdy[c] * invM - xh * sdyx[c] * invM;
      out[e] = stf<T>(weight[c] * invstd[c] * t);
    }
    reinterpret_cast<uint4*>(dx)[v] = *reinterpret_cast<uint4*>(out);
  }
}

template <typename T>
__global__ __launch_bounds__(256) void bn_bwd_apply_scalar(
    const T* __restrict__ x, const T* __restrict__ dy, T* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, const float* __restrict__ sdy,
    const float* __restrict__ sdyx, float invM, size_t total, int C) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    float xh = (ldf(x[i]) - mean[c]) * invstd[c];
    float g = ldf(dy[i]);
    float t = g - sdy[c] * invM - xh * sdyx[c] * invM;
    dx[i] = stf<T>(weight[c] * invstd[c] * t);
  }
}

// Vector-stats geometry: valid when the channel-group count divides the
// block cleanly (all power-of-two C, the ResNet/conv-net case).
struct VecStatsGeom {
  bool ok = false;
  int c8_per_block = 0, c_blocks = 0, row_blocks = 0, grid = 0;
};

VecStatsGeom vec_stats_geom(int C, int VEC, size_t M) {
  VecStatsGeom g;
  if (C % VEC != 0) return g;
  int nC8 = C / VEC;
  int c8pb;
  if (nC8 <= 256 && 256 % nC8 == 0)
    c8pb = nC8;
  else if (nC8 % 256 == 0)
    c8pb = 256;
  else
    return g;
  g.c8_per_block = c8pb;
  g.c_blocks = nC8 / c8pb;
  int rows_per_iter = 256 / c8pb;
  int max_rb = (int)((M + rows_per_iter - 1) / rows_per_iter);
  g.row_blocks = std::min(std::max(2048 / g.c_blocks, 1), std::max(max_rb, 1));
  g.grid = g.c_blocks * g.row_blocks;
  g.ok = true;
  return g;
}

struct BnGeom {
  size_t M;
  int C, c_blocks, row_blocks;
  dim3 stats_grid, stats_block;
  int apply_grid;
};

BnGeom bn_geom(const at::Tensor& x) {
  BnGeom g;
  g.C = (int)x.size(1);
  g.M = (size_t)x.numel() / g.C;
  g.c_blocks = (g.C + kLanes - 1) / kLanes;
  int max_row_blocks = (int)((g.M + kRowsPB - 1) / kRowsPB);
  g.row_blocks = std::min(std::max(2048 / g.c_blocks, 1), max_row_blocks);
  g.stats_grid = dim3(g.c_blocks * g.row_blocks);
  g.stats_block = dim3(kLanes, kRowsPB);
  size_t work = x.numel() / 8 + 1;
  g.apply_grid = (int)std::min<size_t>((work + 255) / 256, 2048);
  return g;
}

void check_bn_input(const at::Tensor& x) {
  TMA_CHECK(x.is_cuda() && x.dim() == 4, "bn: 4D CUDA tensor required");
  TMA_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
            "bn: channels_last layout required");
  TMA_CHECK(x.scalar_type() == at::kBFloat16 || x.scalar_type() == at::kFloat,
            "bn: bf16 or f32 input");
}

}  // namespace

// Returns (y, save_mean, save_invstd).
std::vector<at::Tensor> bn_forward(at::Tensor x, at::Tensor weight, at::Tensor bias,
                                   at::Tensor running_mean, at::Tensor running_var,
                                   bool training, double momentum, double eps) {
  check_bn_input(x);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  BnGeom g = bn_geom(x);
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor mean, invstd;
  if (training) {
    at::Tensor sum = at::zeros({g.C}, fopt);
    at::Tensor sumsq = at::zeros({g.C}, fopt);
    mean = at::empty({g.C}, fopt);
    invstd = at::empty({g.C}, fopt);
    const int vecw = x.scalar_type() == at::kBFloat16 ? 8 : 4;
    VecStatsGeom vg = vec_stats_geom(g.C, vecw, g.M);
    if (vg.ok && x.scalar_type() == at::kBFloat16)
      hipLaunchKernelGGL((bn_stats_vec_kernel<__hip_bfloat16, 8>), dim3(vg.grid),
                         dim3(256), 0, stream,
                         (const __hip_bfloat16*)x.data_ptr(),
                         sum.data_ptr<float>(), sumsq.data_ptr<float>(), g.M,
                         g.C / 8, vg.c8_per_block, vg.row_blocks);
    else if (vg.ok)
      hipLaunchKernelGGL((bn_stats_vec_kernel<float, 4>), dim3(vg.grid),
                         dim3(256), 0, stream, x.data_ptr<float>(),
                         sum.data_ptr<float>(), sumsq.data_ptr<float>(), g.M,
                         g.C / 4, vg.c8_per_block, vg.row_blocks);
    else if (x.scalar_type() == at::kBFloat16)
      hipLaunchKernelGGL(bn_stats_kernel<__hip_bfloat16>, g.stats_grid,
                         g.stats_block, 0, stream,
                         (const __hip_bfloat16*)x.data_ptr(),
                         sum.data_ptr<float>(), sumsq.data_ptr<float>(), g.M, g.C,
                         g.c_blocks, g.row_blocks);
    else
      hipLaunchKernelGGL(bn_stats_kernel<float>, g.stats_grid, g.stats_block, 0,
                         stream, x.data_ptr<float>(), sum.data_ptr<float>(),
                         sumsq.data_ptr<float>(), g.M, g.C, g.c_blocks,
                         g.row_blocks);
    hipLaunchKernelGGL(bn_finalize_kernel, dim3((g.C + 255) / 256), dim3(256), 0,
                       stream, sum.data_ptr<float>(), sumsq.data_ptr<float>(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       running_mean.defined() ? running_mean.data_ptr<float>()
                                              : nullptr,
                       running_var.defined() ? running_var.data_ptr<float>()
                                             : nullptr,
                       (float)momentum, (float)eps, (float)g.M, g.C);
  } else {
    mean = running_mean;
    invstd = (running_var + eps).rsqrt();
  }
  at::Tensor y = at::empty_like(x);
  size_t total = (size_t)x.numel();
  bool vec_ok = (g.C % 8 == 0) && x.scalar_type() == at::kBFloat16;
  bool vec4_ok = (g.C % 4 == 0) && x.scalar_type() == at::kFloat;
  if (x.scalar_type() == at::kBFloat16) {
    if (vec_ok)
      hipLaunchKernelGGL((bn_fwd_apply_kernel<__hip_bfloat16, 8>),
                         dim3(g.apply_grid), dim3(256), 0, stream,
                         (const __hip_bfloat16*)x.data_ptr(),
                         (__hip_bfloat16*)y.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), weight.data_ptr<float>(),
                         bias.data_ptr<float>(), total, g.C);
    else
      hipLaunchKernelGGL(bn_fwd_apply_scalar<__hip_bfloat16>, dim3(g.apply_grid),
                         dim3(256), 0, stream,
                         (const __hip_bfloat16*)x.data_ptr(),
                         (__hip_bfloat16*)y.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), weight.data_ptr<float>(),
                         bias.data_ptr<float>(), total, g.C);
  } else {
    if (vec4_ok)
      hipLaunchKernelGGL((bn_fwd_apply_kernel<float, 4>), dim3(g.apply_grid),
                         dim3(256), 0, stream, x.data_ptr<float>(),
                         y.data_ptr<float>(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), weight.data_ptr<float>(),
                         bias.data_ptr<float>(), total, g.C);
    else
      hipLaunchKernelGGL(bn_fwd_apply_scalar<float>, dim3(g.apply_grid), dim3(256),
                         0, stream, x.data_ptr<float>(), y.data_ptr<float>(),
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         weight.data_ptr<float>(), bias.data_ptr<float>(), total,
                         g.C);
  }
  return {y, mean, invstd};
}

// Returns (dx, dweight, dbias).
std::vector<at::Tensor> bn_backward(at::Tensor x, at::Tensor dy, at::Tensor weight,
                                    at::Tensor mean, at::Tensor invstd) {
  check_bn_input(x);
  check_bn_input(dy);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  BnGeom g = bn_geom(x);
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor sdy = at::zeros({g.C}, fopt);
  at::Tensor sdyx = at::zeros({g.C}, fopt);
  const int vecw = x.scalar_type() == at::kBFloat16 ? 8 : 4;
  VecStatsGeom vg = vec_stats_geom(g.C, vecw, g.M);
  if (vg.ok && x.scalar_type() == at::kBFloat16)
    hipLaunchKernelGGL((bn_bwd_stats_vec_kernel<__hip_bfloat16, 8>),
                       dim3(vg.grid), dim3(256), 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(),
                       (const __hip_bfloat16*)dy.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       sdy.data_ptr<float>(), sdyx.data_ptr<float>(), g.M,
                       g.C / 8, vg.c8_per_block, vg.row_blocks);
  else if (vg.ok)
    hipLaunchKernelGGL((bn_bwd_stats_vec_kernel<float, 4>), dim3(vg.grid),
                       dim3(256), 0, stream, x.data_ptr<float>(),
                       dy.data_ptr<float>(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), sdy.data_ptr<float>(),
                       sdyx.data_ptr<float>(), g.M, g.C / 4, vg.c8_per_block,
                       vg.row_blocks);
  else if (x.scalar_type() == at::kBFloat16)
    hipLaunchKernelGGL(bn_bwd_stats_kernel<__hip_bfloat16>, g.stats_grid,
                       g.stats_block, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(),
                       (const __hip_bfloat16*)dy.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       sdy.data_ptr<float>(), sdyx.data_ptr<float>(), g.M, g.C,
                       g.c_blocks, g.row_blocks);
  else
    hipLaunchKernelGGL(bn_bwd_stats_kernel<float>, g.stats_grid, g.stats_block, 0,
                       stream, x.data_ptr<float>(), dy.data_ptr<float>(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       sdy.data_ptr<float>(), sdyx.data_ptr<float>(), g.M, g.C,
                       g.c_blocks, g.row_blocks);
  at::Tensor dx = at::empty_like(x);
  size_t total = (size_t)x.numel();
  float invM = 1.f / (float)g.M;
  if (x.scalar_type() == at::kBFloat16) {
    if (g.C % 8 == 0)
      hipLaunchKernelGGL((bn_bwd_apply_kernel<__hip_bfloat16, 8>),
                         dim3(g.apply_grid), dim3(256), 0, stream,
                         (const __hip_bfloat16*)x.data_ptr(),
                         (const __hip_bfloat16*)dy.data_ptr(),
                         (__hip_bfloat16*)dx.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), weight.data_ptr<float>(),
                         sdy.data_ptr<float>(), sdyx.data_ptr<float>(), invM,
                         total, g.C);
    else
      hipLaunchKernelGGL(bn_bwd_apply_scalar<__hip_bfloat16>, dim3(g.apply_grid),
                         dim3(256), 0, stream,
                         (const __hip_bfloat16*)x.data_ptr(),
                         (const __hip_bfloat16*)dy.data_ptr(),
                         (__hip_bfloat16*)dx.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), weight.data_ptr<float>(),
                         sdy.data_ptr<float>(), sdyx.data_ptr<float>(), invM,
                         total, g.C);
  } else {
    if (g.C % 4 == 0)
      hipLaunchKernelGGL((bn_bwd_apply_kernel<float, 4>), dim3(g.apply_grid),
                         dim3(256), 0, stream, x.data_ptr<float>(),
                         dy.data_ptr<float>(), dx.data_ptr<float>(),
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         weight.data_ptr<float>(), sdy.data_ptr<float>(),
                         sdyx.data_ptr<float>(), invM, total, g.C);
    else
      hipLaunchKernelGGL(bn_bwd_apply_scalar<float>, dim3(g.apply_grid), dim3(256),
                         0, stream, x.data_ptr<float>(), dy.data_ptr<float>(),
                         dx.data_ptr<float>(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), weight.data_ptr<float>(),
                         sdy.data_ptr<float>(), sdyx.data_ptr<float>(), invM,
                         total, g.C);
  }
  return {dx, sdyx, sdy};  // dweight = sum(dy*xhat), dbias = sum(dy)
}

void register_batchnorm(pybind11::module_& m) {
  m.def("bn_forward", &bn_forward);
  m.def("bn_backward", &bn_backward);
}

}  // namespace tma
