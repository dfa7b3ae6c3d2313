// Fused NHWC (channels_last) BatchNorm for gfx950 — training fwd + bwd.
//
// Motivation (profiles/resnet50_bf16_kernel_stats.md): MIOpen's spatial
// BatchNorm kernels are ~35% of the ResNet-50 bf16 step on MI355X.
//
// Design (v2, after ISA inspection of v1 — see profiles/):
//   - One thread owns a FIXED group of VEC consecutive channels (VEC = 8
//     bf16 / 4 f32 = 16 B) and loops over rows. Because the channel-group
//     count divides the 256-thread block and nC8 * 16 B = one row's bytes,
//     a wave's lanes always cover contiguous memory — full-rate dwordx4
//     streams for any power-of-two C.
//   - Fixed channels mean per-channel parameters (mean/invstd/weight/bias/
//     sums) are loaded ONCE into registers — v1 issued 32 scalar global
//     loads per 16 B of data (L1-thrashed, latency-bound, 0.3 TB/s).
//   - Reductions write per-(row-block) partials (plain stores) reduced by
//     the finalize kernel — v1's global atomicAdd chains were depth
//     row_blocks (=2048) and cost a flat ~420 us regardless of size.
// Stats and affine params are fp32; x/y/dy may be bf16 or f32.
#include <hip/hip_runtime.h>
#include <cstdlib>
#include <hip/hip_bf16.h>
#include <hip/hip_cooperative_groups.h>

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

// Geometry shared by all vector kernels: block = (c8_per_block channel
// groups) x (256 / c8_per_block row copies); grid = c_blocks x row_blocks.
struct VecGeom {
  bool ok = false;
  int nC8 = 0, c8_per_block = 0, c_blocks = 0, row_blocks = 0, grid = 0;
};

inline int bn_rb_cap() {
  static int cap = -1;
  if (cap < 0) {
    const char* v = std::getenv("TORCHMPI_BN_RB");
    cap = v ? std::max(64, atoi(v)) : 512;  // measured sweep, see profiles/
  }
  return cap;
}

VecGeom vec_geom(int C, int VEC, size_t M) {
  VecGeom g;
  if (C % VEC != 0) return g;
  g.nC8 = C / VEC;
  if (g.nC8 <= 256 && 256 % g.nC8 == 0)
    g.c8_per_block = g.nC8;
  else if (g.nC8 % 256 == 0)
    g.c8_per_block = 256;
  else
    return g;
  g.c_blocks = g.nC8 / g.c8_per_block;
  int rows_per_iter = 256 / g.c8_per_block;
  int max_rb = (int)((M + rows_per_iter - 1) / rows_per_iter);
  g.row_blocks = std::min(std::max(bn_rb_cap() / g.c_blocks, 1), std::max(max_rb, 1));
  // Small-M/large-C shapes: keep the partial buffers (row_blocks * C * 8 B)
  // under half the data read, or the reduction traffic dominates.
  int traffic_cap = std::max(1, (int)(M / VEC));
  g.row_blocks = std::min(g.row_blocks, traffic_cap);
  g.grid = g.c_blocks * g.row_blocks;
  g.ok = true;
  return g;
}

// ---- vector stats: per-(row_block, channel) partial sums -----------------

template <typename T, int VEC>
__global__ __launch_bounds__(256) void bn_stats_vec_kernel(
    const T* __restrict__ x, float* __restrict__ psum, float* __restrict__ psumsq,
    size_t M, int nC8, int c8_per_block, int row_blocks) {
  const int c_blocks = nC8 / c8_per_block;
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int rows_per_iter = 256 / c8_per_block;
  const int c8l = (int)threadIdx.x % c8_per_block;
  const int rsub = (int)threadIdx.x / c8_per_block;
  const int c8 = cb * c8_per_block + c8l;
  const int C = nC8 * VEC;
  float acc[VEC], accsq[VEC];
#pragma unroll
  for (int e = 0; e < VEC; ++e) acc[e] = accsq[e] = 0.f;
  const size_t row_stride = (size_t)row_blocks * rows_per_iter;
  const T* base = x + (size_t)c8 * VEC;
  for (size_t r = (size_t)rb * rows_per_iter + rsub; r < M; r += row_stride) {
    T v[VEC];
    *reinterpret_cast<uint4*>(v) =
        *reinterpret_cast<const uint4*>(base + r * (size_t)C);
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      float f = ldf(v[e]);
      acc[e] += f;
      accsq[e] += f * f;
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
  for (int e = 0; e < VEC; ++e) ls[threadIdx.x * VEC + e] = accsq[e];
  __syncthreads();
  if (rsub == 0) {
    for (int w = 1; w < rows_per_iter; ++w)
#pragma unroll
      for (int e = 0; e < VEC; ++e)
        accsq[e] += ls[(w * c8_per_block + c8l) * VEC + e];
    float* ps = psum + (size_t)rb * C + c8 * VEC;
    float* pq = psumsq + (size_t)rb * C + c8 * VEC;
#pragma unroll
    for (int e = 0; e < VEC; e += 4) {
      *reinterpret_cast<float4*>(ps + e) =
          make_float4(acc[e], acc[e + 1], acc[e + 2], acc[e + 3]);
      *reinterpret_cast<float4*>(pq + e) =
          make_float4(accsq[e], accsq[e + 1], accsq[e + 2], accsq[e + 3]);
    }
  }
}

// Scalar fallback (odd C): atomics into sum/sumsq directly — cold path.
template <typename T>
__global__ __launch_bounds__(256) void bn_stats_scalar_kernel(
    const T* __restrict__ x, float* __restrict__ sum, float* __restrict__ sumsq,
    size_t M, int C, int c_blocks, int row_blocks) {
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int lane = (int)threadIdx.x % 64;
  const int rsub = (int)threadIdx.x / 64;
  const int c = cb * 64 + lane;
  float s = 0.f, ss = 0.f;
  if (c < C) {
    const size_t row_stride = (size_t)row_blocks * 4;
    for (size_t r = (size_t)rb * 4 + rsub; r < M; r += row_stride) {
      float v = ldf(x[r * C + c]);
      s += v;
      ss += v * v;
    }
  }
  __shared__ float ls[4][64];
  __shared__ float lss[4][64];
  ls[rsub][lane] = s;
  lss[rsub][lane] = ss;
  __syncthreads();
  if (rsub == 0 && c < C) {
    float ts = 0.f, tss = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      ts += ls[w][lane];
      tss += lss[w][lane];
    }
    atomicAdd(&sum[c], ts);
    atomicAdd(&sumsq[c], tss);
  }
}

// ---- finalize: reduce partials + mean/invstd + running stats -------------

// One block per channel; 256 threads cooperatively reduce the row_blocks
// partials (a single flat-indexed block was 1.2 ms at row_blocks=4096).
__global__ __launch_bounds__(256) void bn_finalize_kernel(
    const float* __restrict__ psum, const float* __restrict__ psumsq,
    int row_blocks, float* __restrict__ mean, float* __restrict__ invstd,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    float momentum, float eps, float M, int C) {
  const int c = blockIdx.x;
  float s = 0.f, ss = 0.f;
  for (int rb = threadIdx.x; rb < row_blocks; rb += 256) {
    s += psum[(size_t)rb * C + c];
    ss += psumsq[(size_t)rb * C + c];
  }
  __shared__ float l1[256], l2[256];
  l1[threadIdx.x] = s;
  l2[threadIdx.x] = ss;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if ((int)threadIdx.x < w) {
      l1[threadIdx.x] += l1[threadIdx.x + w];
      l2[threadIdx.x] += l2[threadIdx.x + w];
    }
    __syncthreads();
  }
  if (threadIdx.x != 0) return;
  s = l1[0];
  ss = l2[0];
  float m = s / M;
  float var = fmaxf(ss / M - m * m, 0.f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    float unbiased = M > 1.f ? var * M / (M - 1.f) : var;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

__global__ __launch_bounds__(256) void bn_bwd_finalize_kernel(
    const float* __restrict__ psdy, const float* __restrict__ psdyx,
    int row_blocks, float* __restrict__ sdy, float* __restrict__ sdyx, int C) {
  const int c = blockIdx.x;
  float a = 0.f, b = 0.f;
  for (int rb = threadIdx.x; rb < row_blocks; rb += 256) {
    a += psdy[(size_t)rb * C + c];
    b += psdyx[(size_t)rb * C + c];
  }
  __shared__ float l1[256], l2[256];
  l1[threadIdx.x] = a;
  l2[threadIdx.x] = b;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if ((int)threadIdx.x < w) {
      l1[threadIdx.x] += l1[threadIdx.x + w];
      l2[threadIdx.x] += l2[threadIdx.x + w];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    sdy[c] = l1[0];
    sdyx[c] = l2[0];
  }
}

// ---- forward apply (fixed channels per thread, params in registers) ------
// RELU / RES fuse the following activation and residual add into the BN
// epilogue: y = [relu](sc*x + sh [+ res]) — saves the separate elementwise
// passes that were ~20% of the ResNet-50 step (profiles/
// resnet50_bf16_kernel_stats.md).

template <typename T, int VEC, bool RELU, bool RES>
__global__ __launch_bounds__(256) void bn_fwd_apply_vec(
    const T* __restrict__ x, const T* __restrict__ res, T* __restrict__ y,
    uint8_t* __restrict__ mask, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ weight,
    const float* __restrict__ bias, size_t M, int nC8, int c8_per_block,
    int row_blocks) {
  const int c_blocks = nC8 / c8_per_block;
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int rows_per_iter = 256 / c8_per_block;
  const int c8l = (int)threadIdx.x % c8_per_block;
  const int rsub = (int)threadIdx.x / c8_per_block;
  const int c8 = cb * c8_per_block + c8l;
  const int C = nC8 * VEC;
  float sc[VEC], sh[VEC];
#pragma unroll
  for (int e = 0; e < VEC; ++e) {
    int c = c8 * VEC + e;
    sc[e] = weight[c] * invstd[c];  // y = sc * x + sh
    sh[e] = bias[c] - mean[c] * sc[e];
  }
  const size_t row_stride = (size_t)row_blocks * rows_per_iter;
  const T* xb = x + (size_t)c8 * VEC;
  const T* rb2 = RES ? res + (size_t)c8 * VEC : nullptr;
  T* yb = y + (size_t)c8 * VEC;
  for (size_t r = (size_t)rb * rows_per_iter + rsub; r < M; r += row_stride) {
    T v[VEC], o[VEC], rv[VEC];
    *reinterpret_cast<uint4*>(v) =
        *reinterpret_cast<const uint4*>(xb + r * (size_t)C);
    if (RES)
      *reinterpret_cast<uint4*>(rv) =
          *reinterpret_cast<const uint4*>(rb2 + r * (size_t)C);
    unsigned mb = 0;
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      float f = ldf(v[e]) * sc[e] + sh[e];
      if (RES) f += ldf(rv[e]);
      if (RELU) {
        mb |= (unsigned)(f > 0.f) << e;
        f = fmaxf(f, 0.f);
      }
      o[e] = stf<T>(f);
    }
    *reinterpret_cast<uint4*>(yb + r * (size_t)C) = *reinterpret_cast<uint4*>(o);
    // 1-bit relu mask, one byte per VEC elements: backward reads this
    // instead of re-streaming y (saves ~2 activation passes per step).
    if (RELU) mask[r * (size_t)nC8 + c8] = (uint8_t)mb;
  }
}

template <typename T, bool RELU, bool RES>
__global__ __launch_bounds__(256) void bn_fwd_apply_scalar(
    const T* __restrict__ x, const T* __restrict__ res, T* __restrict__ y,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, const float* __restrict__ bias,
    size_t total, int C) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    float xh = (ldf(x[i]) - mean[c]) * invstd[c];
    float f = xh * weight[c] + bias[c];
    if (RES) f += ldf(res[i]);
    if (RELU) f = fmaxf(f, 0.f);
    y[i] = stf<T>(f);
  }
}

// ---- backward stats ------------------------------------------------------

template <typename T, int VEC, bool RELU>
__global__ __launch_bounds__(256) void bn_bwd_stats_vec_kernel(
    const T* __restrict__ x, const T* __restrict__ dy,
    const uint8_t* __restrict__ mask, const float* __restrict__ mean,
    const float* __restrict__ invstd, float* __restrict__ psdy,
    float* __restrict__ psdyx, size_t M, int nC8, int c8_per_block,
    int row_blocks) {
  const int c_blocks = nC8 / c8_per_block;
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int rows_per_iter = 256 / c8_per_block;
  const int c8l = (int)threadIdx.x % c8_per_block;
  const int rsub = (int)threadIdx.x / c8_per_block;
  const int c8 = cb * c8_per_block + c8l;
  const int C = nC8 * VEC;
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
        *reinterpret_cast<const uint4*>(xb + r * (size_t)C);
    *reinterpret_cast<uint4*>(gv) =
        *reinterpret_cast<const uint4*>(gb + r * (size_t)C);
    unsigned mb = RELU ? mask[r * (size_t)nC8 + c8] : 0u;
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      float g = ldf(gv[e]);
      if (RELU && !((mb >> e) & 1u)) g = 0.f;  // relu mask from forward
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
    float* pa = psdy + (size_t)rb * C + c8 * VEC;
    float* pb = psdyx + (size_t)rb * C + c8 * VEC;
#pragma unroll
    for (int e = 0; e < VEC; e += 4) {
      *reinterpret_cast<float4*>(pa + e) =
          make_float4(acc[e], acc[e + 1], acc[e + 2], acc[e + 3]);
      *reinterpret_cast<float4*>(pb + e) =
          make_float4(accx[e], accx[e + 1], accx[e + 2], accx[e + 3]);
    }
  }
}

template <typename T, bool RELU>
__global__ __launch_bounds__(256) void bn_bwd_stats_scalar_kernel(
    const T* __restrict__ x, const T* __restrict__ dy,
    const T* __restrict__ yout, const float* __restrict__ mean,
    const float* __restrict__ invstd, float* __restrict__ sdy,
    float* __restrict__ sdyx, size_t M, int C, int c_blocks, int row_blocks) {
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int lane = (int)threadIdx.x % 64;
  const int rsub = (int)threadIdx.x / 64;
  const int c = cb * 64 + lane;
  float s = 0.f, sx = 0.f;
  if (c < C) {
    const float mu = mean[c], is = invstd[c];
    const size_t row_stride = (size_t)row_blocks * 4;
    for (size_t r = (size_t)rb * 4 + rsub; r < M; r += row_stride) {
      float g = ldf(dy[r * C + c]);
      if (RELU && ldf(yout[r * C + c]) <= 0.f) g = 0.f;
      float xh = (ldf(x[r * C + c]) - mu) * is;
      s += g;
      sx += g * xh;
    }
  }
  __shared__ float ls[4][64];
  __shared__ float lsx[4][64];
  ls[rsub][lane] = s;
  lsx[rsub][lane] = sx;
  __syncthreads();
  if (rsub == 0 && c < C) {
    float ts = 0.f, tsx = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      ts += ls[w][lane];
      tsx += lsx[w][lane];
    }
    atomicAdd(&sdy[c], ts);
    atomicAdd(&sdyx[c], tsx);
  }
}

// ---- backward apply ------------------------------------------------------

// RELU: dy is masked by (yout > 0) everywhere it appears; RES: the residual
// branch's gradient dres = masked dy is produced in the same pass.
template <typename T, int VEC, bool RELU, bool RES>
__global__ __launch_bounds__(256) void bn_bwd_apply_vec(
    const T* __restrict__ x, const T* __restrict__ dy,
    const uint8_t* __restrict__ mask, T* __restrict__ dx, T* __restrict__ dres,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, const float* __restrict__ sdy,
    const float* __restrict__ sdyx, float invM, size_t M, int nC8,
    int c8_per_block, int row_blocks) {
  const int c_blocks = nC8 / c8_per_block;
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int rows_per_iter = 256 / c8_per_block;
  const int c8l = (int)threadIdx.x % c8_per_block;
  const int rsub = (int)threadIdx.x / c8_per_block;
  const int c8 = cb * c8_per_block + c8l;
  const int C = nC8 * VEC;
  // dx = w*is*(g - sdy/M - xhat*sdyx/M) = A*g + B*x + D  (g = masked dy)
  float A[VEC], B[VEC], D[VEC];
#pragma unroll
  for (int e = 0; e < VEC; ++e) {
    int c = c8 * VEC + e;
    float a = weight[c] * invstd[c];
    float b = -a * invstd[c] * sdyx[c] * invM;
    A[e] = a;
    B[e] = b;
    D[e] = -a * sdy[c] * invM - b * mean[c];
  }
  const size_t row_stride = (size_t)row_blocks * rows_per_iter;
  const T* xb = x + (size_t)c8 * VEC;
  const T* gb = dy + (size_t)c8 * VEC;
  T* ob = dx + (size_t)c8 * VEC;
  T* eb = RES ? dres + (size_t)c8 * VEC : nullptr;
  for (size_t r = (size_t)rb * rows_per_iter + rsub; r < M; r += row_stride) {
    T xv[VEC], gv[VEC], o[VEC], ev[VEC];
    *reinterpret_cast<uint4*>(xv) =
        *reinterpret_cast<const uint4*>(xb + r * (size_t)C);
    *reinterpret_cast<uint4*>(gv) =
        *reinterpret_cast<const uint4*>(gb + r * (size_t)C);
    unsigned mb = RELU ? mask[r * (size_t)nC8 + c8] : 0u;
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      float g = ldf(gv[e]);
      if (RELU && !((mb >> e) & 1u)) g = 0.f;
      o[e] = stf<T>(A[e] * g + B[e] * ldf(xv[e]) + D[e]);
      if (RES) ev[e] = stf<T>(g);
    }
    *reinterpret_cast<uint4*>(ob + r * (size_t)C) = *reinterpret_cast<uint4*>(o);
    if (RES)
      *reinterpret_cast<uint4*>(eb + r * (size_t)C) =
          *reinterpret_cast<uint4*>(ev);
  }
}

template <typename T, bool RELU, bool RES>
__global__ __launch_bounds__(256) void bn_bwd_apply_scalar(
    const T* __restrict__ x, const T* __restrict__ dy,
    const T* __restrict__ yout, T* __restrict__ dx, T* __restrict__ dres,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, const float* __restrict__ sdy,
    const float* __restrict__ sdyx, float invM, size_t total, int C) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    float xh = (ldf(x[i]) - mean[c]) * invstd[c];
    float g = ldf(dy[i]);
    if (RELU && ldf(yout[i]) <= 0.f) g = 0.f;
    float t = g - sdy[c] * invM - xh * sdyx[c] * invM;
    dx[i] = stf<T>(weight[c] * invstd[c] * t);
    if (RES) dres[i] = stf<T>(g);
  }
}

// ---- cooperative single-kernel BN (training fwd / bwd) --------------------
// One hipLaunchCooperativeKernel replaces the stats -> finalize -> apply
// 3-kernel chain: grid.sync() separates the phases, and the finalize runs
// with the FULL grid's parallelism (the dedicated finalize kernel had only
// C blocks and cost ~2.4 ms/step in launch/ramp — profiles/
// resnet50_r2_fusion.md). Two-level finalize: slices of row_blocks reduce
// into ps2/pq2 [nsl x C], then one thread per channel folds the slices.

namespace cg = cooperative_groups;

template <typename T, int VEC, bool RELU, bool RES>
__global__ __launch_bounds__(256) void bn_fwd_coop_kernel(
    const T* __restrict__ x, const T* __restrict__ res, T* __restrict__ y,
    uint8_t* __restrict__ mask, float* __restrict__ psum,
    float* __restrict__ psumsq, float* __restrict__ ps2,
    float* __restrict__ pq2, float* __restrict__ mean,
    float* __restrict__ invstd, float* __restrict__ running_mean,
    float* __restrict__ running_var, const float* __restrict__ weight,
    const float* __restrict__ bias, float momentum, float eps, size_t M,
    int nC8, int c8_per_block, int row_blocks, int nsl) {
  cg::grid_group grid = cg::this_grid();
  const int c_blocks = nC8 / c8_per_block;
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int rows_per_iter = 256 / c8_per_block;
  const int c8l = (int)threadIdx.x % c8_per_block;
  const int rsub = (int)threadIdx.x / c8_per_block;
  const int c8 = cb * c8_per_block + c8l;
  const int C = nC8 * VEC;

  // ---- phase 1: per-(row-block, channel) partial sums
  {
    float acc[VEC], accsq[VEC];
#pragma unroll
    for (int e = 0; e < VEC; ++e) acc[e] = accsq[e] = 0.f;
    const size_t row_stride = (size_t)row_blocks * rows_per_iter;
    const T* base = x + (size_t)c8 * VEC;
    for (size_t r = (size_t)rb * rows_per_iter + rsub; r < M; r += row_stride) {
      T v[VEC];
      *reinterpret_cast<uint4*>(v) =
          *reinterpret_cast<const uint4*>(base + r * (size_t)C);
#pragma unroll
      for (int e = 0; e < VEC; ++e) {
        float f = ldf(v[e]);
        acc[e] += f;
        accsq[e] += f * f;
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
    for (int e = 0; e < VEC; ++e) ls[threadIdx.x * VEC + e] = accsq[e];
    __syncthreads();
    if (rsub == 0) {
      for (int w = 1; w < rows_per_iter; ++w)
#pragma unroll
        for (int e = 0; e < VEC; ++e)
          accsq[e] += ls[(w * c8_per_block + c8l) * VEC + e];
      float* ps = psum + (size_t)rb * C + c8 * VEC;
      float* pq = psumsq + (size_t)rb * C + c8 * VEC;
#pragma unroll
      for (int e = 0; e < VEC; e += 4) {
        *reinterpret_cast<float4*>(ps + e) =
            make_float4(acc[e], acc[e + 1], acc[e + 2], acc[e + 3]);
        *reinterpret_cast<float4*>(pq + e) =
            make_float4(accsq[e], accsq[e + 1], accsq[e + 2], accsq[e + 3]);
      }
    }
  }
  grid.sync();

  // ---- phase 2a: slice-reduce partials (full grid parallelism)
  {
    const size_t T_ = (size_t)gridDim.x * 256;
    const size_t gtid = (size_t)blockIdx.x * 256 + threadIdx.x;
    const int span = (row_blocks + nsl - 1) / nsl;
    for (size_t w = gtid; w < (size_t)C * nsl; w += T_) {
      int c = (int)(w % C);
      int s = (int)(w / C);
      int r0 = s * span, r1 = min(r0 + span, row_blocks);
      float a = 0.f, b = 0.f;
      for (int r = r0; r < r1; ++r) {
        a += psum[(size_t)r * C + c];
        b += psumsq[(size_t)r * C + c];
      }
      ps2[(size_t)s * C + c] = a;
      pq2[(size_t)s * C + c] = b;
    }
  }
  grid.sync();

  // ---- phase 2b: per-channel mean/invstd (+ running stats)
  {
    const size_t T_ = (size_t)gridDim.x * 256;
    const size_t gtid = (size_t)blockIdx.x * 256 + threadIdx.x;
    const float Mf = (float)M;
    for (size_t c = gtid; c < (size_t)C; c += T_) {
      float s = 0.f, ss = 0.f;
      for (int k = 0; k < nsl; ++k) {
        s += ps2[(size_t)k * C + c];
        ss += pq2[(size_t)k * C + c];
      }
      float m = s / Mf;
      float var = fmaxf(ss / Mf - m * m, 0.f);
      mean[c] = m;
      invstd[c] = rsqrtf(var + eps);
      if (running_mean) {
        running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
        float unbiased = Mf > 1.f ? var * Mf / (Mf - 1.f) : var;
        running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
      }
    }
  }
  grid.sync();

  // ---- phase 3: apply (y = [relu](sc*x + sh [+ res]); mask bits)
  {
    float sc[VEC], sh[VEC];
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      int c = c8 * VEC + e;
      sc[e] = weight[c] * invstd[c];
      sh[e] = bias[c] - mean[c] * sc[e];
    }
    const size_t row_stride = (size_t)row_blocks * rows_per_iter;
    const T* xb = x + (size_t)c8 * VEC;
    const T* rb2 = RES ? res + (size_t)c8 * VEC : nullptr;
    T* yb = y + (size_t)c8 * VEC;
    for (size_t r = (size_t)rb * rows_per_iter + rsub; r < M; r += row_stride) {
      T v[VEC], o[VEC], rv[VEC];
      *reinterpret_cast<uint4*>(v) =
          *reinterpret_cast<const uint4*>(xb + r * (size_t)C);
      if (RES)
        *reinterpret_cast<uint4*>(rv) =
            *reinterpret_cast<const uint4*>(rb2 + r * (size_t)C);
      unsigned mb = 0;
#pragma unroll
      for (int e = 0; e < VEC; ++e) {
        float f = ldf(v[e]) * sc[e] + sh[e];
        if (RES) f += ldf(rv[e]);
        if (RELU) {
          mb |= (unsigned)(f > 0.f) << e;
          f = fmaxf(f, 0.f);
        }
        o[e] = stf<T>(f);
      }
      *reinterpret_cast<uint4*>(yb + r * (size_t)C) = *reinterpret_cast<uint4*>(o);
      if (RELU) mask[r * (size_t)nC8 + c8] = (uint8_t)mb;
    }
  }
}

template <typename T, int VEC, bool RELU, bool RES>
__global__ __launch_bounds__(256) void bn_bwd_coop_kernel(
    const T* __restrict__ x, const T* __restrict__ dy,
    const uint8_t* __restrict__ mask, T* __restrict__ dx,
    T* __restrict__ dres, float* __restrict__ psdy, float* __restrict__ psdyx,
    float* __restrict__ ps2, float* __restrict__ pq2,
    float* __restrict__ sdy, float* __restrict__ sdyx,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, float invM, size_t M, int nC8,
    int c8_per_block, int row_blocks, int nsl) {
  cg::grid_group grid = cg::this_grid();
  const int c_blocks = nC8 / c8_per_block;
  const int cb = blockIdx.x % c_blocks;
  const int rb = blockIdx.x / c_blocks;
  const int rows_per_iter = 256 / c8_per_block;
  const int c8l = (int)threadIdx.x % c8_per_block;
  const int rsub = (int)threadIdx.x / c8_per_block;
  const int c8 = cb * c8_per_block + c8l;
  const int C = nC8 * VEC;

  // ---- phase 1: partial sdy / sdyx (relu-masked)
  {
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
          *reinterpret_cast<const uint4*>(xb + r * (size_t)C);
      *reinterpret_cast<uint4*>(gv) =
          *reinterpret_cast<const uint4*>(gb + r * (size_t)C);
      unsigned mb = RELU ? mask[r * (size_t)nC8 + c8] : 0u;
#pragma unroll
      for (int e = 0; e < VEC; ++e) {
        float g = ldf(gv[e]);
        if (RELU && !((mb >> e) & 1u)) g = 0.f;
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
      float* pa = psdy + (size_t)rb * C + c8 * VEC;
      float* pb = psdyx + (size_t)rb * C + c8 * VEC;
#pragma unroll
      for (int e = 0; e < VEC; e += 4) {
        *reinterpret_cast<float4*>(pa + e) =
            make_float4(acc[e], acc[e + 1], acc[e + 2], acc[e + 3]);
        *reinterpret_cast<float4*>(pb + e) =
            make_float4(accx[e], accx[e + 1], accx[e + 2], accx[e + 3]);
      }
    }
  }
  grid.sync();

  // ---- phase 2a: slice reduce
  {
    const size_t T_ = (size_t)gridDim.x * 256;
    const size_t gtid = (size_t)blockIdx.x * 256 + threadIdx.x;
    const int span = (row_blocks + nsl - 1) / nsl;
    for (size_t w = gtid; w < (size_t)C * nsl; w += T_) {
      int c = (int)(w % C);
      int s = (int)(w / C);
      int r0 = s * span, r1 = min(r0 + span, row_blocks);
      float a = 0.f, b = 0.f;
      for (int r = r0; r < r1; ++r) {
        a += psdy[(size_t)r * C + c];
        b += psdyx[(size_t)r * C + c];
      }
      ps2[(size_t)s * C + c] = a;
      pq2[(size_t)s * C + c] = b;
    }
  }
  grid.sync();

  // ---- phase 2b: final per-channel sums
  {
    const size_t T_ = (size_t)gridDim.x * 256;
    const size_t gtid = (size_t)blockIdx.x * 256 + threadIdx.x;
    for (size_t c = gtid; c < (size_t)C; c += T_) {
      float a = 0.f, b = 0.f;
      for (int k = 0; k < nsl; ++k) {
        a += ps2[(size_t)k * C + c];
        b += pq2[(size_t)k * C + c];
      }
      sdy[c] = a;
      sdyx[c] = b;
    }
  }
  grid.sync();

  // ---- phase 3: apply (dx = A*g + B*x + D; dres = g)
  {
    float A[VEC], B[VEC], D[VEC];
#pragma unroll
    for (int e = 0; e < VEC; ++e) {
      int c = c8 * VEC + e;
      float a = weight[c] * invstd[c];
      float b = -a * invstd[c] * sdyx[c] * invM;
      A[e] = a;
      B[e] = b;
      D[e] = -a * sdy[c] * invM - b * mean[c];
    }
    const size_t row_stride = (size_t)row_blocks * rows_per_iter;
    const T* xb = x + (size_t)c8 * VEC;
    const T* gb = dy + (size_t)c8 * VEC;
    T* ob = dx + (size_t)c8 * VEC;
    T* eb = RES ? dres + (size_t)c8 * VEC : nullptr;
    for (size_t r = (size_t)rb * rows_per_iter + rsub; r < M; r += row_stride) {
      T xv[VEC], gv[VEC], o[VEC], ev[VEC];
      *reinterpret_cast<uint4*>(xv) =
          *reinterpret_cast<const uint4*>(xb + r * (size_t)C);
      *reinterpret_cast<uint4*>(gv) =
          *reinterpret_cast<const uint4*>(gb + r * (size_t)C);
      unsigned mb = RELU ? mask[r * (size_t)nC8 + c8] : 0u;
#pragma unroll
      for (int e = 0; e < VEC; ++e) {
        float g = ldf(gv[e]);
        if (RELU && !((mb >> e) & 1u)) g = 0.f;
        o[e] = stf<T>(A[e] * g + B[e] * ldf(xv[e]) + D[e]);
        if (RES) ev[e] = stf<T>(g);
      }
      *reinterpret_cast<uint4*>(ob + r * (size_t)C) = *reinterpret_cast<uint4*>(o);
      if (RES)
        *reinterpret_cast<uint4*>(eb + r * (size_t)C) =
            *reinterpret_cast<uint4*>(ev);
    }
  }
}

// Max co-resident blocks for a cooperative kernel (cached per instantiation);
// -1 = cooperative launch unavailable.
template <typename KernT>
int coop_max_blocks(KernT kern) {
  static int cached = -2;
  if (cached == -2) {
    // MEASURED OFF by default: on ROCm 7.2 the cooperative launch path cost
    // far more than the finalize it fuses away (bn_bench 0.2-0.4x, end-to-end
    // 8099 -> 3121 samples/s; gpurun_out/r7_*). Numerics verified; opt in
    // with TORCHMPI_BN_COOP=1 to re-evaluate on newer stacks.
    const char* on = std::getenv("TORCHMPI_BN_COOP");
    if (!(on && std::string(on) == "1")) {
      cached = -1;
      return cached;
    }
    int dev = 0;
    if (hipGetDevice(&dev) != hipSuccess) return (cached = -1);
    hipDeviceProp_t prop;
    if (hipGetDeviceProperties(&prop, dev) != hipSuccess) return (cached = -1);
    if (!prop.cooperativeLaunch) return (cached = -1);
    int nb = 0;
    hipError_t e = hipOccupancyMaxActiveBlocksPerMultiprocessor(&nb, kern, 256, 0);
    cached = (e == hipSuccess && nb > 0) ? nb * prop.multiProcessorCount : -1;
  }
  return cached;
}

void check_bn_input(const at::Tensor& x) {
  TMA_CHECK(x.is_cuda() && x.dim() == 4, "bn: 4D CUDA tensor required");
  TMA_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
            "bn: channels_last layout required");
  TMA_CHECK(x.scalar_type() == at::kBFloat16 || x.scalar_type() == at::kFloat,
            "bn: bf16 or f32 input");
}

int scalar_grid(size_t work) {
  size_t blocks = (work + 255) / 256;
  return (int)std::min<size_t>(blocks ? blocks : 1, 2048);
}

}  // namespace

// Returns (y, save_mean, save_invstd). relu/res fuse the activation and
// residual add into the apply epilogue (y = [relu](bn(x) [+ res])).
std::vector<at::Tensor> bn_forward(at::Tensor x, at::Tensor weight, at::Tensor bias,
                                   at::Tensor running_mean, at::Tensor running_var,
                                   bool training, double momentum, double eps,
                                   bool relu, c10::optional<at::Tensor> res_opt) {
  check_bn_input(x);
  at::Tensor res = res_opt.value_or(at::Tensor());
  const bool has_res = res.defined();
  if (has_res) {
    TMA_CHECK(res.sizes() == x.sizes() && res.scalar_type() == x.scalar_type(),
              "bn: residual shape/dtype mismatch");
    TMA_CHECK(res.is_contiguous(at::MemoryFormat::ChannelsLast),
              "bn: residual must be channels_last");
  }
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const int C = (int)x.size(1);
  const size_t M = (size_t)x.numel() / C;
  const bool bf16 = x.scalar_type() == at::kBFloat16;
  const int vecw = bf16 ? 8 : 4;
  VecGeom vg = vec_geom(C, vecw, M);
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor mean, invstd;

  // Cooperative single-kernel fast path: stats -> finalize -> apply in one
  // launch with grid.sync (falls back to the 3-kernel chain whenever
  // cooperative launch is unavailable or the grid exceeds residency).
  if (training && vg.ok) {
    mean = at::empty({C}, fopt);
    invstd = at::empty({C}, fopt);
    at::Tensor y = at::empty_like(x);
    at::Tensor mask;
    if (relu)
      mask = at::empty({(int64_t)(M * (size_t)vg.nC8)},
                       at::TensorOptions().dtype(at::kByte).device(x.device()));
    bool done = false;
#define TMA_BN_FWD_COOP(T, VEC)                                                \
  do {                                                                         \
    auto try_coop = [&](auto kern) -> bool {                                   \
      int maxb = coop_max_blocks(kern);                                        \
      if (maxb <= 0 || vg.c_blocks > maxb) return false;                       \
      int rbk = std::min(vg.row_blocks, std::max(1, maxb / vg.c_blocks));      \
      int grid = vg.c_blocks * rbk;                                            \
      int nsl = std::min(64, rbk);                                             \
      at::Tensor psum = at::empty({(int64_t)rbk * C}, fopt);                   \
      at::Tensor psumsq = at::empty({(int64_t)rbk * C}, fopt);                 \
      at::Tensor ps2 = at::empty({(int64_t)nsl * C}, fopt);                    \
      at::Tensor pq2 = at::empty({(int64_t)nsl * C}, fopt);                    \
      const T* xp = (const T*)x.data_ptr();                                    \
      const T* rp = has_res ? (const T*)res.data_ptr() : nullptr;              \
      T* yp = (T*)y.data_ptr();                                                \
      uint8_t* mkp = mask.defined() ? mask.data_ptr<uint8_t>() : nullptr;      \
      float* psp = psum.data_ptr<float>();                                     \
      float* pqp = psumsq.data_ptr<float>();                                   \
      float* p2 = ps2.data_ptr<float>();                                       \
      float* q2 = pq2.data_ptr<float>();                                       \
      float* mnp = mean.data_ptr<float>();                                     \
      float* ivp = invstd.data_ptr<float>();                                   \
      float* rmp = running_mean.defined() ? running_mean.data_ptr<float>()     \
                                          : nullptr;                          \
      float* rvp = running_var.defined() ? running_var.data_ptr<float>()       \
                                         : nullptr;                          \
      const float* wp = weight.data_ptr<float>();                             \
      const float* bp = bias.data_ptr<float>();                               \
      float mom = (float)momentum, ep = (float)eps;                           \
      size_t Mv = M;                                                          \
      int nc8 = vg.nC8, c8pb = vg.c8_per_block;                               \
      void* kargs[] = {&xp, &rp, &yp, &mkp, &psp, &pqp, &p2,  &q2,            \
                       &mnp, &ivp, &rmp, &rvp, &wp, &bp, &mom, &ep,           \
                       &Mv, &nc8, &c8pb, &rbk, &nsl};                         \
      hipError_t e = hipLaunchCooperativeKernel(                              \
          reinterpret_cast<const void*>(kern), dim3(grid), dim3(256), kargs,  \
          0, stream);                                                         \
      if (e != hipSuccess) {                                                  \
        (void)hipGetLastError();                                              \
        return false;                                                         \
      }                                                                       \
      return true;                                                            \
    };                                                                        \
    if (relu && has_res) done = try_coop(bn_fwd_coop_kernel<T, VEC, true, true>);   \
    else if (relu) done = try_coop(bn_fwd_coop_kernel<T, VEC, true, false>);  \
    else if (has_res) done = try_coop(bn_fwd_coop_kernel<T, VEC, false, true>);     \
    else done = try_coop(bn_fwd_coop_kernel<T, VEC, false, false>);           \
  } while (0)
    if (bf16)
      TMA_BN_FWD_COOP(__hip_bfloat16, 8);
    else
      TMA_BN_FWD_COOP(float, 4);
#undef TMA_BN_FWD_COOP
    if (done) {
      if (!mask.defined())
        mask = at::empty({0},
                         at::TensorOptions().dtype(at::kByte).device(x.device()));
      return {y, mean, invstd, mask};
    }
  }

  if (training) {
    mean = at::empty({C}, fopt);
    invstd = at::empty({C}, fopt);
    if (vg.ok) {
      at::Tensor psum = at::empty({(int64_t)vg.row_blocks * C}, fopt);
      at::Tensor psumsq = at::empty({(int64_t)vg.row_blocks * C}, fopt);
      if (bf16)
        hipLaunchKernelGGL((bn_stats_vec_kernel<__hip_bfloat16, 8>),
                           dim3(vg.grid), dim3(256), 0, stream,
                           (const __hip_bfloat16*)x.data_ptr(),
                           psum.data_ptr<float>(), psumsq.data_ptr<float>(), M,
                           vg.nC8, vg.c8_per_block, vg.row_blocks);
      else
        hipLaunchKernelGGL((bn_stats_vec_kernel<float, 4>), dim3(vg.grid),
                           dim3(256), 0, stream, x.data_ptr<float>(),
                           psum.data_ptr<float>(), psumsq.data_ptr<float>(), M,
                           vg.nC8, vg.c8_per_block, vg.row_blocks);
      hipLaunchKernelGGL(bn_finalize_kernel, dim3(C), dim3(256), 0,
                         stream, psum.data_ptr<float>(), psumsq.data_ptr<float>(),
                         vg.row_blocks, mean.data_ptr<float>(),
                         invstd.data_ptr<float>(),
                         running_mean.defined() ? running_mean.data_ptr<float>()
                                                : nullptr,
                         running_var.defined() ? running_var.data_ptr<float>()
                                               : nullptr,
                         (float)momentum, (float)eps, (float)M, C);
    } else {
      at::Tensor sum = at::zeros({C}, fopt);
      at::Tensor sumsq = at::zeros({C}, fopt);
      int c_blocks = (C + 63) / 64;
      int row_blocks = std::max(1, 512 / c_blocks);
      dim3 grid(c_blocks * row_blocks), block(256);
      if (bf16)
        hipLaunchKernelGGL(bn_stats_scalar_kernel<__hip_bfloat16>, grid, block, 0,
                           stream, (const __hip_bfloat16*)x.data_ptr(),
                           sum.data_ptr<float>(), sumsq.data_ptr<float>(), M, C,
                           c_blocks, row_blocks);
      else
        hipLaunchKernelGGL(bn_stats_scalar_kernel<float>, grid, block, 0, stream,
                           x.data_ptr<float>(), sum.data_ptr<float>(),
                           sumsq.data_ptr<float>(), M, C, c_blocks, row_blocks);
      hipLaunchKernelGGL(bn_finalize_kernel, dim3(C), dim3(256), 0,
                         stream, sum.data_ptr<float>(), sumsq.data_ptr<float>(),
                         1, mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         running_mean.defined() ? running_mean.data_ptr<float>()
                                                : nullptr,
                         running_var.defined() ? running_var.data_ptr<float>()
                                               : nullptr,
                         (float)momentum, (float)eps, (float)M, C);
    }
  } else {
    mean = running_mean.contiguous();
    invstd = (running_var + eps).rsqrt();
  }

  at::Tensor y = at::empty_like(x);
  // 1-bit relu mask (one byte per VEC elements) for the backward kernels —
  // replaces re-streaming y; only the vector path uses it.
  at::Tensor mask;
  if (relu && vg.ok)
    mask = at::empty({(int64_t)(M * (size_t)vg.nC8)},
                     at::TensorOptions().dtype(at::kByte).device(x.device()));
#define TMA_BN_FWD(T, VEC)                                                     \
  do {                                                                         \
    const T* xp = (const T*)x.data_ptr();                                      \
    const T* rp = has_res ? (const T*)res.data_ptr() : nullptr;                \
    T* yp = (T*)y.data_ptr();                                                  \
    uint8_t* mp = mask.defined() ? mask.data_ptr<uint8_t>() : nullptr;         \
    if (vg.ok) {                                                               \
      auto args = [&](auto kern) {                                             \
        hipLaunchKernelGGL(kern, dim3(vg.grid), dim3(256), 0, stream, xp, rp,  \
                           yp, mp, mean.data_ptr<float>(),                     \
                           invstd.data_ptr<float>(), weight.data_ptr<float>(), \
                           bias.data_ptr<float>(), M, vg.nC8, vg.c8_per_block, \
                           vg.row_blocks);                                     \
      };                                                                       \
      if (relu && has_res) args(bn_fwd_apply_vec<T, VEC, true, true>);         \
      else if (relu) args(bn_fwd_apply_vec<T, VEC, true, false>);              \
      else if (has_res) args(bn_fwd_apply_vec<T, VEC, false, true>);           \
      else args(bn_fwd_apply_vec<T, VEC, false, false>);                       \
    } else {                                                                   \
      size_t total = (size_t)x.numel();                                        \
      auto args = [&](auto kern) {                                             \
        hipLaunchKernelGGL(kern, dim3(scalar_grid(total)), dim3(256), 0,       \
                           stream, xp, rp, yp, mean.data_ptr<float>(),         \
                           invstd.data_ptr<float>(), weight.data_ptr<float>(), \
                           bias.data_ptr<float>(), total, C);                  \
      };                                                                       \
      if (relu && has_res) args(bn_fwd_apply_scalar<T, true, true>);           \
      else if (relu) args(bn_fwd_apply_scalar<T, true, false>);                \
      else if (has_res) args(bn_fwd_apply_scalar<T, false, true>);             \
      else args(bn_fwd_apply_scalar<T, false, false>);                         \
    }                                                                          \
  } while (0)
  if (bf16)
    TMA_BN_FWD(__hip_bfloat16, 8);
  else
    TMA_BN_FWD(float, 4);
#undef TMA_BN_FWD
  if (!mask.defined())
    mask = at::empty({0}, at::TensorOptions().dtype(at::kByte).device(x.device()));
  return {y, mean, invstd, mask};
}

// Returns (dx, dweight, dbias[, dres]). `yout` is the fused forward output
// (needed for the relu mask); `need_dres` additionally emits the residual
// branch's gradient (= relu-masked dy) in the same pass.
std::vector<at::Tensor> bn_backward(at::Tensor x, at::Tensor dy, at::Tensor weight,
                                    at::Tensor mean, at::Tensor invstd,
                                    c10::optional<at::Tensor> yout_opt, bool relu,
                                    bool need_dres,
                                    c10::optional<at::Tensor> mask_opt) {
  at::Tensor yout = yout_opt.value_or(at::Tensor());
  at::Tensor mask = mask_opt.value_or(at::Tensor());
  if (mask.defined() && mask.numel() == 0) mask = at::Tensor();
  check_bn_input(x);
  check_bn_input(dy);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const int C = (int)x.size(1);
  const size_t M = (size_t)x.numel() / C;
  const bool bf16 = x.scalar_type() == at::kBFloat16;
  const int vecw = bf16 ? 8 : 4;
  VecGeom vg = vec_geom(C, vecw, M);
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor sdy = at::empty({C}, fopt);
  at::Tensor sdyx = at::empty({C}, fopt);
  at::Tensor dx = at::empty_like(x);
  at::Tensor dres;
  if (need_dres) dres = at::empty_like(x);
  float invM = 1.f / (float)M;

  // Cooperative single-kernel fast path (mirrors forward).
  if (vg.ok && (!relu || mask.defined())) {
    bool done = false;
#define TMA_BN_BWD_COOP(T, VEC)                                                \
  do {                                                                         \
    auto try_coop = [&](auto kern) -> bool {                                   \
      int maxb = coop_max_blocks(kern);                                        \
      if (maxb <= 0 || vg.c_blocks > maxb) return false;                       \
      int rbk = std::min(vg.row_blocks, std::max(1, maxb / vg.c_blocks));      \
      int grid = vg.c_blocks * rbk;                                            \
      int nsl = std::min(64, rbk);                                             \
      at::Tensor psdy = at::empty({(int64_t)rbk * C}, fopt);                   \
      at::Tensor psdyx = at::empty({(int64_t)rbk * C}, fopt);                  \
      at::Tensor ps2 = at::empty({(int64_t)nsl * C}, fopt);                    \
      at::Tensor pq2 = at::empty({(int64_t)nsl * C}, fopt);                    \
      const T* xp = (const T*)x.data_ptr();                                    \
      const T* gp = (const T*)dy.data_ptr();                                   \
      const uint8_t* mkp =                                                     \
          mask.defined() ? mask.data_ptr<uint8_t>() : nullptr;                 \
      T* dxp = (T*)dx.data_ptr();                                              \
      T* drp = need_dres ? (T*)dres.data_ptr() : nullptr;                      \
      float* pa = psdy.data_ptr<float>();                                      \
      float* pb = psdyx.data_ptr<float>();                                     \
      float* p2 = ps2.data_ptr<float>();                                       \
      float* q2 = pq2.data_ptr<float>();                                       \
      float* sa = sdy.data_ptr<float>();                                       \
      float* sb = sdyx.data_ptr<float>();                                      \
      const float* mnp = mean.data_ptr<float>();                               \
      const float* ivp = invstd.data_ptr<float>();                             \
      const float* wp = weight.data_ptr<float>();                              \
      float im = invM;                                                         \
      size_t Mv = M;                                                           \
      int nc8 = vg.nC8, c8pb = vg.c8_per_block;                                \
      void* kargs[] = {&xp, &gp, &mkp, &dxp, &drp, &pa, &pb, &p2, &q2,         \
                       &sa, &sb, &mnp, &ivp, &wp, &im, &Mv, &nc8, &c8pb,       \
                       &rbk, &nsl};                                            \
      hipError_t e = hipLaunchCooperativeKernel(                               \
          reinterpret_cast<const void*>(kern), dim3(grid), dim3(256), kargs,   \
          0, stream);                                                          \
      if (e != hipSuccess) {                                                   \
        (void)hipGetLastError();                                               \
        return false;                                                          \
      }                                                                        \
      return true;                                                             \
    };                                                                         \
    if (relu && need_dres)                                                     \
      done = try_coop(bn_bwd_coop_kernel<T, VEC, true, true>);                 \
    else if (relu) done = try_coop(bn_bwd_coop_kernel<T, VEC, true, false>);   \
    else if (need_dres)                                                        \
      done = try_coop(bn_bwd_coop_kernel<T, VEC, false, true>);                \
    else done = try_coop(bn_bwd_coop_kernel<T, VEC, false, false>);            \
  } while (0)
    if (bf16)
      TMA_BN_BWD_COOP(__hip_bfloat16, 8);
    else
      TMA_BN_BWD_COOP(float, 4);
#undef TMA_BN_BWD_COOP
    if (done) {
      if (need_dres) return {dx, sdyx, sdy, dres};
      return {dx, sdyx, sdy};
    }
  }

#define TMA_BN_BWD(T, VEC)                                                     \
  do {                                                                         \
    const T* xp = (const T*)x.data_ptr();                                      \
    const T* gp = (const T*)dy.data_ptr();                                     \
    const T* yp = (relu && yout.defined()) ? (const T*)yout.data_ptr()         \
                                           : nullptr;                          \
    const uint8_t* mp =                                                        \
        (relu && mask.defined()) ? mask.data_ptr<uint8_t>() : nullptr;         \
    T* dxp = (T*)dx.data_ptr();                                                \
    T* drp = need_dres ? (T*)dres.data_ptr() : nullptr;                        \
    if (vg.ok) {                                                               \
      TMA_CHECK(!relu || mp, "bn_backward: relu vec path needs the mask");     \
      at::Tensor psdy = at::empty({(int64_t)vg.row_blocks * C}, fopt);         \
      at::Tensor psdyx = at::empty({(int64_t)vg.row_blocks * C}, fopt);        \
      auto sargs = [&](auto kern) {                                            \
        hipLaunchKernelGGL(kern, dim3(vg.grid), dim3(256), 0, stream, xp, gp,  \
                           mp, mean.data_ptr<float>(),                         \
                           invstd.data_ptr<float>(), psdy.data_ptr<float>(),   \
                           psdyx.data_ptr<float>(), M, vg.nC8,                 \
                           vg.c8_per_block, vg.row_blocks);                    \
      };                                                                       \
      if (relu) sargs(bn_bwd_stats_vec_kernel<T, VEC, true>);                  \
      else sargs(bn_bwd_stats_vec_kernel<T, VEC, false>);                      \
      hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3(C), dim3(256), 0,        \
                         stream, psdy.data_ptr<float>(),                       \
                         psdyx.data_ptr<float>(), vg.row_blocks,               \
                         sdy.data_ptr<float>(), sdyx.data_ptr<float>(), C);    \
      auto aargs = [&](auto kern) {                                            \
        hipLaunchKernelGGL(kern, dim3(vg.grid), dim3(256), 0, stream, xp, gp,  \
                           mp, dxp, drp, mean.data_ptr<float>(),               \
                           invstd.data_ptr<float>(), weight.data_ptr<float>(), \
                           sdy.data_ptr<float>(), sdyx.data_ptr<float>(),      \
                           invM, M, vg.nC8, vg.c8_per_block, vg.row_blocks);   \
      };                                                                       \
      if (relu && need_dres) aargs(bn_bwd_apply_vec<T, VEC, true, true>);      \
      else if (relu) aargs(bn_bwd_apply_vec<T, VEC, true, false>);             \
      else if (need_dres) aargs(bn_bwd_apply_vec<T, VEC, false, true>);        \
      else aargs(bn_bwd_apply_vec<T, VEC, false, false>);                      \
    } else {                                                                   \
      TMA_CHECK(!relu || yp, "bn_backward: relu scalar path needs yout");      \
      sdy.zero_();                                                             \
      sdyx.zero_();                                                            \
      int c_blocks = (C + 63) / 64;                                            \
      int row_blocks = std::max(1, 512 / c_blocks);                            \
      dim3 grid(c_blocks * row_blocks), block(256);                            \
      auto sargs = [&](auto kern) {                                            \
        hipLaunchKernelGGL(kern, grid, block, 0, stream, xp, gp, yp,           \
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),   \
                           sdy.data_ptr<float>(), sdyx.data_ptr<float>(), M,   \
                           C, c_blocks, row_blocks);                           \
      };                                                                       \
      if (relu) sargs(bn_bwd_stats_scalar_kernel<T, true>);                    \
      else sargs(bn_bwd_stats_scalar_kernel<T, false>);                        \
      size_t total = (size_t)x.numel();                                        \
      auto aargs = [&](auto kern) {                                            \
        hipLaunchKernelGGL(kern, dim3(scalar_grid(total)), dim3(256), 0,       \
                           stream, xp, gp, yp, dxp, drp,                       \
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),   \
                           weight.data_ptr<float>(), sdy.data_ptr<float>(),    \
                           sdyx.data_ptr<float>(), invM, total, C);            \
      };                                                                       \
      if (relu && need_dres) aargs(bn_bwd_apply_scalar<T, true, true>);        \
      else if (relu) aargs(bn_bwd_apply_scalar<T, true, false>);               \
      else if (need_dres) aargs(bn_bwd_apply_scalar<T, false, true>);          \
      else aargs(bn_bwd_apply_scalar<T, false, false>);                        \
    }                                                                          \
  } while (0)
  if (bf16)
    TMA_BN_BWD(__hip_bfloat16, 8);
  else
    TMA_BN_BWD(float, 4);
#undef TMA_BN_BWD
  if (need_dres)
    return {dx, sdyx, sdy, dres};  // dweight = sum(g*xhat), dbias = sum(g)
  return {dx, sdyx, sdy};
}

void register_batchnorm(pybind11::module_& m) {
  namespace py = pybind11;
  m.def("bn_forward", &bn_forward, py::arg("x"), py::arg("weight"),
        py::arg("bias"), py::arg("running_mean"), py::arg("running_var"),
        py::arg("training"), py::arg("momentum"), py::arg("eps"),
        py::arg("relu") = false, py::arg("res") = py::none());
  m.def("bn_backward", &bn_backward, py::arg("x"), py::arg("dy"),
        py::arg("weight"), py::arg("mean"), py::arg("invstd"),
        py::arg("yout") = py::none(), py::arg("relu") = false,
        py::arg("need_dres") = false, py::arg("mask") = py::none());
}

}  // namespace tma
