// Fused RMSNorm (forward + backward) for gfx950 — the Llama norm.
//
// y[r,d] = x[r,d] * inv_r * w[d],  inv_r = rsqrt(mean_d(x^2) + eps)
//
// Eager PyTorch RMSNorm (models/llama.py) is ~7 activation passes per call
// (cast-up, pow, mean, rsqrt, two muls, cast-down); this is 1 read + 1
// write: one 256-thread block per row, x held in registers between the
// sumsq reduction and the scale (D <= 16384 bf16 fits 16 B/lane x 4).
//
// Backward:
//   dx[r,d] = inv_r * w[d] * dy[r,d] - x[r,d] * inv_r^3 / D * s_r,
//     s_r = sum_d dy[r,d] * w[d] * x[r,d]          (row kernel)
//   dw[d]  = sum_r dy[r,d] * x[r,d] * inv_r        (column kernel: fixed
//     feature group per thread + per-row-block partials, like batchnorm.hip)
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

__device__ __forceinline__ float bf2f(__hip_bfloat16 v) {
  return __bfloat162float(v);
}

// Block-wide sum of one float via LDS tree (256 threads).
__device__ __forceinline__ float block_sum(float v, float* lds) {
  lds[threadIdx.x] = v;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if ((int)threadIdx.x < w) lds[threadIdx.x] += lds[threadIdx.x + w];
    __syncthreads();
  }
  float r = lds[0];
  __syncthreads();
  return r;
}

// ---- forward: one block per row; x stays in registers ---------------------
// D must be a multiple of 8 and <= 256 * PERT * 8. PERT = elems/8 per thread.
// ADD fuses the Llama pre-norm residual add: h = x + res (bf16-rounded,
// written to hout and used for the stats/scale so backward sees exactly the
// forward values), y = rmsnorm(h) — saves the separate add pass and the
// norm's re-read of the sum.
template <int PERT, bool ADD>
__global__ __launch_bounds__(256) void rms_fwd_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ res,
    __hip_bfloat16* __restrict__ hout, __hip_bfloat16* __restrict__ y,
    const float* __restrict__ w, float* __restrict__ inv_out, int D, float eps) {
  const size_t row = blockIdx.x;
  const __hip_bfloat16* xr = x + row * (size_t)D;
  const __hip_bfloat16* rr = ADD ? res + row * (size_t)D : nullptr;
  __hip_bfloat16* hr = ADD ? hout + row * (size_t)D : nullptr;
  __hip_bfloat16* yr = y + row * (size_t)D;
  __hip_bfloat16 v[PERT][8];
  float ss = 0.f;
  const int nvec = D / 8;
#pragma unroll
  for (int k = 0; k < PERT; ++k) {
    int i = (int)threadIdx.x + k * 256;
    if (i < nvec) {
      *reinterpret_cast<uint4*>(v[k]) = reinterpret_cast<const uint4*>(xr)[i];
      if (ADD) {
        __hip_bfloat16 rv[8];
        *reinterpret_cast<uint4*>(rv) = reinterpret_cast<const uint4*>(rr)[i];
#pragma unroll
        for (int e = 0; e < 8; ++e)
          v[k][e] = __float2bfloat16(bf2f(v[k][e]) + bf2f(rv[e]));
        reinterpret_cast<uint4*>(hr)[i] = *reinterpret_cast<uint4*>(v[k]);
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float f = bf2f(v[k][e]);
        ss += f * f;
      }
    }
  }
  __shared__ float lds[256];
  ss = block_sum(ss, lds);
  const float inv = rsqrtf(ss / (float)D + eps);
  if (threadIdx.x == 0 && inv_out) inv_out[row] = inv;
#pragma unroll
  for (int k = 0; k < PERT; ++k) {
    int i = (int)threadIdx.x + k * 256;
    if (i < nvec) {
      __hip_bfloat16 o[8];
      const float* wd = w + i * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        o[e] = __float2bfloat16(bf2f(v[k][e]) * inv * wd[e]);
      reinterpret_cast<uint4*>(yr)[i] = *reinterpret_cast<uint4*>(o);
    }
  }
}

// ---- backward dx: one block per row ---------------------------------------
// ADD: dres (the downstream gradient of the residual sum h) is added into
// dx in the same pass — the fused op's both inputs receive this tensor.
template <int PERT, bool ADD>
__global__ __launch_bounds__(256) void rms_bwd_dx_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ dy,
    const __hip_bfloat16* __restrict__ dres, __hip_bfloat16* __restrict__ dx,
    const float* __restrict__ w, const float* __restrict__ inv_in, int D) {
  const size_t row = blockIdx.x;
  const __hip_bfloat16* xr = x + row * (size_t)D;
  const __hip_bfloat16* gr = dy + row * (size_t)D;
  const __hip_bfloat16* er = ADD ? dres + row * (size_t)D : nullptr;
  __hip_bfloat16* or_ = dx + row * (size_t)D;
  const float inv = inv_in[row];
  __hip_bfloat16 xv[PERT][8], gv[PERT][8];
  float s = 0.f;
  const int nvec = D / 8;
#pragma unroll
  for (int k = 0; k < PERT; ++k) {
    int i = (int)threadIdx.x + k * 256;
    if (i < nvec) {
      *reinterpret_cast<uint4*>(xv[k]) = reinterpret_cast<const uint4*>(xr)[i];
      *reinterpret_cast<uint4*>(gv[k]) = reinterpret_cast<const uint4*>(gr)[i];
      const float* wd = w + i * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        s += bf2f(gv[k][e]) * wd[e] * bf2f(xv[k][e]);
    }
  }
  __shared__ float lds[256];
  s = block_sum(s, lds);
  const float c = inv * inv * inv * s / (float)D;
#pragma unroll
  for (int k = 0; k < PERT; ++k) {
    int i = (int)threadIdx.x + k * 256;
    if (i < nvec) {
      __hip_bfloat16 o[8], ev[8];
      if (ADD)
        *reinterpret_cast<uint4*>(ev) = reinterpret_cast<const uint4*>(er)[i];
      const float* wd = w + i * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float g = inv * wd[e] * bf2f(gv[k][e]) - bf2f(xv[k][e]) * c;
        if (ADD) g += bf2f(ev[e]);
        o[e] = __float2bfloat16(g);
      }
      reinterpret_cast<uint4*>(or_)[i] = *reinterpret_cast<uint4*>(o);
    }
  }
}

// ---- backward dw: fixed feature-group per thread, per-row-block partials --
__global__ __launch_bounds__(256) void rms_bwd_dw_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ dy,
    const float* __restrict__ inv_in, float* __restrict__ pdw, size_t R, int nD8,
    int d8_per_block, int row_blocks) {
  const int d_blocks = nD8 / d8_per_block;
  const int db = blockIdx.x % d_blocks;
  const int rb = blockIdx.x / d_blocks;
  const int rows_per_iter = 256 / d8_per_block;
  const int d8l = (int)threadIdx.x % d8_per_block;
  const int rsub = (int)threadIdx.x / d8_per_block;
  const int d8 = db * d8_per_block + d8l;
  const int D = nD8 * 8;
  float acc[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) acc[e] = 0.f;
  const size_t row_stride = (size_t)row_blocks * rows_per_iter;
  for (size_t r = (size_t)rb * rows_per_iter + rsub; r < R; r += row_stride) {
    const float inv = inv_in[r];
    __hip_bfloat16 xv[8], gv[8];
    *reinterpret_cast<uint4*>(xv) =
        *reinterpret_cast<const uint4*>(x + r * (size_t)D + d8 * 8);
    *reinterpret_cast<uint4*>(gv) =
        *reinterpret_cast<const uint4*>(dy + r * (size_t)D + d8 * 8);
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] += bf2f(gv[e]) * bf2f(xv[e]) * inv;
  }
  __shared__ float ls[256 * 8];
#pragma unroll
  for (int e = 0; e < 8; ++e) ls[threadIdx.x * 8 + e] = acc[e];
  __syncthreads();
  if (rsub == 0) {
    for (int w = 1; w < rows_per_iter; ++w)
#pragma unroll
      for (int e = 0; e < 8; ++e)
        acc[e] += ls[(w * d8_per_block + d8l) * 8 + e];
    float* pd = pdw + (size_t)rb * D + d8 * 8;
#pragma unroll
    for (int e = 0; e < 8; e += 4)
      *reinterpret_cast<float4*>(pd + e) =
          make_float4(acc[e], acc[e + 1], acc[e + 2], acc[e + 3]);
  }
}

__global__ __launch_bounds__(256) void rms_dw_finalize_kernel(
    const float* __restrict__ pdw, int row_blocks, float* __restrict__ dw,
    int D) {
  const int d = blockIdx.x;
  float a = 0.f;
  for (int rb = threadIdx.x; rb < row_blocks; rb += 256)
    a += pdw[(size_t)rb * D + d];
  __shared__ float l1[256];
  l1[threadIdx.x] = a;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if ((int)threadIdx.x < w) l1[threadIdx.x] += l1[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) dw[d] = l1[0];
}

bool rms_supported(const at::Tensor& x, int D) {
  return x.is_cuda() && x.scalar_type() == at::kBFloat16 && D % 8 == 0 &&
         D <= 256 * 4 * 8 && x.is_contiguous();
}

}  // namespace

bool rms_ok(at::Tensor x) {
  return rms_supported(x, (int)x.size(-1));
}

// Returns (y, inv) with inv the per-row rsqrt (saved for backward).
std::vector<at::Tensor> rms_forward(at::Tensor x, at::Tensor weight, double eps) {
  const int D = (int)x.size(-1);
  TMA_CHECK(rms_supported(x, D), "rms_forward: unsupported input");
  TMA_CHECK(weight.scalar_type() == at::kFloat && weight.is_contiguous(),
            "rms_forward: fp32 contiguous weight");
  const size_t R = (size_t)x.numel() / D;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  at::Tensor y = at::empty_like(x);
  at::Tensor inv = at::empty({(int64_t)R},
                             at::TensorOptions().dtype(at::kFloat).device(x.device()));
  const int nvec = D / 8;
  const int pert = (nvec + 255) / 256;
#define TMA_RMS_FWD(P)                                                        \
  hipLaunchKernelGGL((rms_fwd_kernel<P, false>), dim3((uint32_t)R), dim3(256), \
                     0, stream, (const __hip_bfloat16*)x.data_ptr(), nullptr,  \
                     nullptr, (__hip_bfloat16*)y.data_ptr(),                   \
                     weight.data_ptr<float>(), inv.data_ptr<float>(), D,       \
                     (float)eps)
  switch (pert) {
    case 1: TMA_RMS_FWD(1); break;
    case 2: TMA_RMS_FWD(2); break;
    case 3: TMA_RMS_FWD(3); break;
    default: TMA_RMS_FWD(4); break;
  }
#undef TMA_RMS_FWD
  return {y, inv};
}

// Fused h = x + res; y = rmsnorm(h). Returns (y, h, inv) — h is the new
// residual stream (bf16-rounded exactly as used for the stats).
std::vector<at::Tensor> rms_forward_add(at::Tensor x, at::Tensor res,
                                        at::Tensor weight, double eps) {
  const int D = (int)x.size(-1);
  TMA_CHECK(rms_supported(x, D), "rms_forward_add: unsupported input");
  TMA_CHECK(res.sizes() == x.sizes() && res.scalar_type() == x.scalar_type() &&
                res.is_contiguous(),
            "rms_forward_add: residual mismatch");
  TMA_CHECK(weight.scalar_type() == at::kFloat && weight.is_contiguous(),
            "rms_forward_add: fp32 contiguous weight");
  const size_t R = (size_t)x.numel() / D;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  at::Tensor y = at::empty_like(x);
  at::Tensor h = at::empty_like(x);
  at::Tensor inv = at::empty({(int64_t)R},
                             at::TensorOptions().dtype(at::kFloat).device(x.device()));
  const int nvec = D / 8;
  const int pert = (nvec + 255) / 256;
#define TMA_RMS_FWDA(P)                                                        \
  hipLaunchKernelGGL((rms_fwd_kernel<P, true>), dim3((uint32_t)R), dim3(256),  \
                     0, stream, (const __hip_bfloat16*)x.data_ptr(),           \
                     (const __hip_bfloat16*)res.data_ptr(),                    \
                     (__hip_bfloat16*)h.data_ptr(),                            \
                     (__hip_bfloat16*)y.data_ptr(), weight.data_ptr<float>(),  \
                     inv.data_ptr<float>(), D, (float)eps)
  switch (pert) {
    case 1: TMA_RMS_FWDA(1); break;
    case 2: TMA_RMS_FWDA(2); break;
    case 3: TMA_RMS_FWDA(3); break;
    default: TMA_RMS_FWDA(4); break;
  }
#undef TMA_RMS_FWDA
  return {y, h, inv};
}

// Returns (dx, dweight[fp32]). `dres` (optional) is the downstream gradient
// of the residual sum h — fused into dx so both inputs of the fused add+norm
// receive one tensor.
std::vector<at::Tensor> rms_backward(at::Tensor x, at::Tensor dy, at::Tensor weight,
                                     at::Tensor inv,
                                     c10::optional<at::Tensor> dres_opt) {
  const int D = (int)x.size(-1);
  TMA_CHECK(rms_supported(x, D) && dy.is_contiguous(),
            "rms_backward: unsupported input");
  at::Tensor dres = dres_opt.value_or(at::Tensor());
  const bool add = dres.defined();
  if (add)
    TMA_CHECK(dres.sizes() == x.sizes() && dres.is_contiguous() &&
                  dres.scalar_type() == x.scalar_type(),
              "rms_backward: dres mismatch");
  const size_t R = (size_t)x.numel() / D;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  at::Tensor dx = at::empty_like(x);
  const int nvec = D / 8;
  const int pert = (nvec + 255) / 256;
#define TMA_RMS_DX(P, A)                                                       \
  hipLaunchKernelGGL((rms_bwd_dx_kernel<P, A>), dim3((uint32_t)R), dim3(256),  \
                     0, stream, (const __hip_bfloat16*)x.data_ptr(),           \
                     (const __hip_bfloat16*)dy.data_ptr(),                     \
                     add ? (const __hip_bfloat16*)dres.data_ptr() : nullptr,   \
                     (__hip_bfloat16*)dx.data_ptr(), weight.data_ptr<float>(), \
                     inv.data_ptr<float>(), D)
  switch (pert * 2 + (add ? 1 : 0)) {
    case 2: TMA_RMS_DX(1, false); break;
    case 3: TMA_RMS_DX(1, true); break;
    case 4: TMA_RMS_DX(2, false); break;
    case 5: TMA_RMS_DX(2, true); break;
    case 6: TMA_RMS_DX(3, false); break;
    case 7: TMA_RMS_DX(3, true); break;
    case 8: TMA_RMS_DX(4, false); break;
    default: TMA_RMS_DX(4, true); break;
  }
#undef TMA_RMS_DX

  // dw: fixed feature groups + partials (batchnorm.hip geometry).
  auto fopt = at::TensorOptions().dtype(at::kFloat).device(x.device());
  const int nD8 = D / 8;
  int d8pb;
  if (nD8 <= 256 && 256 % nD8 == 0)
    d8pb = nD8;
  else if (nD8 % 256 == 0)
    d8pb = 256;
  else
    d8pb = 0;
  at::Tensor dw = at::empty({D}, fopt);
  if (d8pb > 0) {
    const int d_blocks = nD8 / d8pb;
    const int rows_per_iter = 256 / d8pb;
    int max_rb = (int)((R + rows_per_iter - 1) / rows_per_iter);
    int row_blocks = std::min(std::max(2048 / d_blocks, 1), std::max(max_rb, 1));
    at::Tensor pdw = at::empty({(int64_t)row_blocks * D}, fopt);
    hipLaunchKernelGGL(rms_bwd_dw_kernel, dim3(d_blocks * row_blocks), dim3(256),
                       0, stream, (const __hip_bfloat16*)x.data_ptr(),
                       (const __hip_bfloat16*)dy.data_ptr(),
                       inv.data_ptr<float>(), pdw.data_ptr<float>(), R, nD8, d8pb,
                       row_blocks);
    hipLaunchKernelGGL(rms_dw_finalize_kernel, dim3(D), dim3(256), 0, stream,
                       pdw.data_ptr<float>(), row_blocks, dw.data_ptr<float>(), D);
  } else {
    // odd D: eager fallback for the weight grad
    at::Tensor xhat = (x.to(at::kFloat) *
                       inv.view({-1, 1}).to(at::kFloat)).view({(int64_t)R, D});
    dw = (dy.to(at::kFloat).view({(int64_t)R, D}) * xhat).sum(0);
  }
  return {dx, dw};
}

void register_rmsnorm(pybind11::module_& m) {
  namespace py = pybind11;
  m.def("rms_ok", &rms_ok);
  m.def("rms_forward", &rms_forward);
  m.def("rms_forward_add", &rms_forward_add);
  m.def("rms_backward", &rms_backward, py::arg("x"), py::arg("dy"),
        py::arg("weight"), py::arg("inv"), py::arg("dres") = py::none());
}

}  // namespace tma
