// GPU parameter server over xGMI peer mappings (hipIpc) — one-sided.
//
// MI355X-native re-design of TorchMPI's parameter server for GPU tensors
// (lib/parameterserver.cpp, SURVEY.md sections 2.1/3.4). The reference ran a
// server thread per rank over MPI point-to-point; on a fully-connected xGMI
// node the server disappears entirely:
//   - each rank's shard lives in FINE-GRAINED device memory
//     (hipExtMallocWithFlags) exported via hipIpcGetMemHandle (dmabuf mode,
//     HSA_ENABLE_IPC_MODE_LEGACY=0);
//   - every client maps every shard and performs updates/reads with its own
//     kernels over the coherent xGMI fabric:
//       'copy' rule  -> remote vectorized store kernel
//       'add' rule   -> system-scope atomicAdd kernel (correct under
//                       concurrent clients, which the reference serialized
//                       through its server thread)
//       receive      -> remote read kernel into the local tensor.
// Completion is stream semantics (hipEvent on the caller's stream).
// Element-wise atomicity only (a reader may see a half-applied update of
// another client — same staleness class Downpour/EASGD tolerate).
#include <torch/extension.h>

#include <ATen/ATen.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>

#include <memory>
#include <unordered_map>
#include <vector>

#include "common.h"

namespace tma {

#ifndef TMA_HIP_CHECK
#define TMA_HIP_CHECK(cmd)                                                    \
  do {                                                                        \
    hipError_t e = (cmd);                                                     \
    TMA_CHECK(e == hipSuccess, std::string("HIP error: ") + hipGetErrorString(e)); \
  } while (0)
#endif

namespace {

constexpr int kBlock = 256;

__global__ __launch_bounds__(kBlock) void copy_bytes_kernel(
    char* __restrict__ dst, const char* __restrict__ src, size_t n) {
  const size_t nvec = n / 16;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t v = (size_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec;
       v += stride) {
    reinterpret_cast<uint4*>(dst)[v] = reinterpret_cast<const uint4*>(src)[v];
  }
  size_t tail = nvec * 16 + ((size_t)blockIdx.x * blockDim.x + threadIdx.x);
  if (tail < n) dst[tail] = src[tail];
}

// Fallback when src/dst are not mutually 16-byte aligned (e.g. a shard
// offset into the client's full tensor that is not a multiple of 16).
__global__ __launch_bounds__(kBlock) void copy_bytes_scalar_kernel(
    char* __restrict__ dst, const char* __restrict__ src, size_t n) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    dst[i] = src[i];
  }
}

inline void launch_copy_bytes(char* dst, const char* src, size_t n, int grid,
                              hipStream_t stream) {
  bool aligned = ((reinterpret_cast<uintptr_t>(dst) |
                   reinterpret_cast<uintptr_t>(src)) & 15) == 0;
  if (aligned)
    hipLaunchKernelGGL(copy_bytes_kernel, dim3(grid), dim3(kBlock), 0, stream,
                       dst, src, n);
  else
    hipLaunchKernelGGL(copy_bytes_scalar_kernel, dim3(grid), dim3(kBlock), 0,
                       stream, dst, src, n);
}

// System-scope atomic add: correct for concurrent updates from several GPUs
// over the coherent xGMI fabric (shards are fine-grained allocations).
__global__ __launch_bounds__(kBlock) void atomic_add_f32_kernel(
    float* __restrict__ dst, const float* __restrict__ src, size_t n) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    __hip_atomic_fetch_add(dst + i, src[i], __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_SYSTEM);
  }
}

__global__ __launch_bounds__(kBlock) void atomic_add_f64_kernel(
    double* __restrict__ dst, const double* __restrict__ src, size_t n) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    __hip_atomic_fetch_add(dst + i, src[i], __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_SYSTEM);
  }
}

// 16-bit float 'add' via 32-bit system-scope CAS (no native 16-bit global
// atomics across xGMI): each thread owns one aligned word = two elements.
// The shard base is allocation-aligned and its byte size is rounded up to
// 16, so the odd-count tail word is always in bounds.
template <int DT>  // 0 = bf16, 1 = f16
__device__ __forceinline__ float h16_to_f(uint16_t v) {
  if (DT == 0) {
    union { uint32_t u; float f; } c;
    c.u = (uint32_t)v << 16;
    return c.f;
  }
  __half_raw r;
  r.x = v;
  return __half2float(__half(r));
}

template <int DT>
__device__ __forceinline__ uint16_t f_to_h16(float f) {
  if (DT == 0) {
    union { uint32_t u; float f; } c;
    c.f = f;
    uint32_t rnd = 0x7fff + ((c.u >> 16) & 1);  // round-to-nearest-even
    return (uint16_t)((c.u + rnd) >> 16);
  }
  __half_raw r = __half_raw(__float2half(f));
  return r.x;
}

template <int DT>
__global__ __launch_bounds__(kBlock) void atomic_add_16_kernel(
    uint32_t* __restrict__ dstw, const uint16_t* __restrict__ src, size_t n) {
  const size_t nwords = n / 2;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (size_t w = tid; w < nwords; w += stride) {
    float a0 = h16_to_f<DT>(src[2 * w]);
    float a1 = h16_to_f<DT>(src[2 * w + 1]);
    uint32_t old = __hip_atomic_load(dstw + w, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_SYSTEM);
    uint32_t desired;
    do {
      uint16_t lo = (uint16_t)(old & 0xffffu);
      uint16_t hi = (uint16_t)(old >> 16);
      desired = (uint32_t)f_to_h16<DT>(h16_to_f<DT>(lo) + a0) |
                ((uint32_t)f_to_h16<DT>(h16_to_f<DT>(hi) + a1) << 16);
    } while (!__hip_atomic_compare_exchange_strong(
        dstw + w, &old, desired, __ATOMIC_RELAXED, __ATOMIC_RELAXED,
        __HIP_MEMORY_SCOPE_SYSTEM));
  }
  if ((n & 1) && tid == 0) {
    // last element occupies the low half of word `nwords`; preserve the high
    // half (allocation padding, never read).
    float a = h16_to_f<DT>(src[n - 1]);
    uint32_t old = __hip_atomic_load(dstw + nwords, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_SYSTEM);
    uint32_t desired;
    do {
      uint16_t lo = (uint16_t)(old & 0xffffu);
      desired = (old & 0xffff0000u) |
                (uint32_t)f_to_h16<DT>(h16_to_f<DT>(lo) + a);
    } while (!__hip_atomic_compare_exchange_strong(
        dstw + nwords, &old, desired, __ATOMIC_RELAXED, __ATOMIC_RELAXED,
        __HIP_MEMORY_SCOPE_SYSTEM));
  }
}

inline int ps_grid(size_t work) {
  size_t blocks = (work + kBlock - 1) / kBlock;
  if (blocks > 2048) blocks = 2048;
  if (blocks == 0) blocks = 1;
  return (int)blocks;
}

struct IpcReg {
  size_t count = 0;
  DType dtype = DType::F32;
  std::vector<size_t> off, cnt;      // shard layout (elements)
  void* own_shard = nullptr;         // fine-grained allocation (this rank)
  std::vector<void*> peer_shards;    // mapped pointers, [world]
};

}  // namespace

class IpcPs {
 public:
  IpcPs(int rank, int world, int device)
      : rank_(rank), world_(world), device_(device) {
    TMA_HIP_CHECK(hipSetDevice(device_));
  }

  ~IpcPs() {
    for (auto& kv : regs_) release(kv.second);
  }

  // Step 1 (collective): allocate + fill own shard, return its IPC handle.
  pybind11::bytes register_local(int tid, at::Tensor full_init) {
    TMA_CHECK(full_init.is_cuda() && full_init.is_contiguous(),
              "ps: init tensor must be contiguous CUDA");
    IpcReg r;
    r.count = (size_t)full_init.numel();
    r.dtype = at_dtype(full_init);
    split_segments(r.count, world_, r.off, r.cnt);
    size_t es = dtype_size(r.dtype);
    // Round up to 16 B: keeps the 16-bit CAS 'add' tail word and the
    // vectorized copy in bounds for any element count.
    size_t bytes = std::max<size_t>((r.cnt[rank_] * es + 15) & ~size_t(15), 16);
    TMA_HIP_CHECK(hipExtMallocWithFlags(&r.own_shard, bytes,
                                        hipDeviceMallocFinegrained));
    TMA_HIP_CHECK(hipMemcpy(r.own_shard,
                            static_cast<char*>(full_init.data_ptr()) +
                                r.off[rank_] * es,
                            r.cnt[rank_] * es, hipMemcpyDeviceToDevice));
    hipIpcMemHandle_t h;
    TMA_HIP_CHECK(hipIpcGetMemHandle(&h, r.own_shard));
    regs_[tid] = r;
    return pybind11::bytes(reinterpret_cast<char*>(&h), sizeof(h));
  }

  // Step 2 (after exchanging handles): map every peer's shard.
  void open_peers(int tid, std::vector<std::string> handles) {
    auto& r = reg(tid);
    TMA_CHECK((int)handles.size() == world_, "ps: need one handle per rank");
    r.peer_shards.assign(world_, nullptr);
    for (int p = 0; p < world_; ++p) {
      if (p == rank_) {
        r.peer_shards[p] = r.own_shard;
        continue;
      }
      TMA_CHECK(handles[p].size() == sizeof(hipIpcMemHandle_t),
                "ps: bad ipc handle size");
      hipIpcMemHandle_t h;
      std::memcpy(&h, handles[p].data(), sizeof(h));
      TMA_HIP_CHECK(hipIpcOpenMemHandle(&r.peer_shards[p], h,
                                        hipIpcMemLazyEnablePeerAccess));
    }
  }

  // Push full-tensor `data` shard-wise to all owners. rule 0=copy, 1=add.
  void send(int tid, at::Tensor data, int rule) {
    auto& r = reg(tid);
    check(data, r);
    auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
    size_t es = dtype_size(r.dtype);
    char* base = static_cast<char*>(data.data_ptr());
    for (int p = 0; p < world_; ++p) {
      if (r.cnt[p] == 0) continue;
      char* src = base + r.off[p] * es;
      void* dst = r.peer_shards.at(p);
      if (rule == 0) {
        launch_copy_bytes(static_cast<char*>(dst), src, r.cnt[p] * es,
                          ps_grid(r.cnt[p] * es / 16), stream);
      } else if (r.dtype == DType::F32) {
        hipLaunchKernelGGL(atomic_add_f32_kernel, dim3(ps_grid(r.cnt[p])),
                           dim3(kBlock), 0, stream, static_cast<float*>(dst),
                           reinterpret_cast<float*>(src), r.cnt[p]);
      } else if (r.dtype == DType::F64) {
        hipLaunchKernelGGL(atomic_add_f64_kernel, dim3(ps_grid(r.cnt[p])),
                           dim3(kBlock), 0, stream, static_cast<double*>(dst),
                           reinterpret_cast<double*>(src), r.cnt[p]);
      } else if (r.dtype == DType::BF16) {
        hipLaunchKernelGGL(atomic_add_16_kernel<0>, dim3(ps_grid(r.cnt[p] / 2 + 1)),
                           dim3(kBlock), 0, stream, static_cast<uint32_t*>(dst),
                           reinterpret_cast<uint16_t*>(src), r.cnt[p]);
      } else if (r.dtype == DType::F16) {
        hipLaunchKernelGGL(atomic_add_16_kernel<1>, dim3(ps_grid(r.cnt[p] / 2 + 1)),
                           dim3(kBlock), 0, stream, static_cast<uint32_t*>(dst),
                           reinterpret_cast<uint16_t*>(src), r.cnt[p]);
      } else {
        TMA_CHECK(false, "ps GPU 'add' rule supports f32/f64/bf16/f16 shards");
      }
    }
  }

  // Gather all shards into full-tensor `data` (remote reads over xGMI).
  void receive(int tid, at::Tensor data) {
    auto& r = reg(tid);
    check(data, r);
    auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
    size_t es = dtype_size(r.dtype);
    char* base = static_cast<char*>(data.data_ptr());
    for (int p = 0; p < world_; ++p) {
      if (r.cnt[p] == 0) continue;
      launch_copy_bytes(base + r.off[p] * es,
                        static_cast<const char*>(r.peer_shards.at(p)),
                        r.cnt[p] * es, ps_grid(r.cnt[p] * es / 16), stream);
    }
  }

  void free_tensor(int tid) {
    auto it = regs_.find(tid);
    if (it == regs_.end()) return;
    release(it->second);
    regs_.erase(it);
  }

 private:
  DType at_dtype(const at::Tensor& t) {
    switch (t.scalar_type()) {
      case at::kFloat: return DType::F32;
      case at::kDouble: return DType::F64;
      case at::kBFloat16: return DType::BF16;
      case at::kHalf: return DType::F16;
      default: TMA_CHECK(false, "ps: unsupported dtype");
    }
    return DType::F32;
  }

  void check(const at::Tensor& t, const IpcReg& r) {
    TMA_CHECK(t.is_cuda() && t.is_contiguous(), "ps: contiguous CUDA tensor");
    TMA_CHECK((size_t)t.numel() == r.count, "ps: size mismatch");
    TMA_CHECK(at_dtype(t) == r.dtype, "ps: dtype mismatch");
    TMA_CHECK(!r.peer_shards.empty(), "ps: open_peers not called");
  }

  IpcReg& reg(int tid) {
    auto it = regs_.find(tid);
    TMA_CHECK(it != regs_.end(), "ps: unregistered tensor id");
    return it->second;
  }

  void release(IpcReg& r) {
    for (int p = 0; p < world_ && p < (int)r.peer_shards.size(); ++p) {
      if (p != rank_ && r.peer_shards[p]) {
        (void)hipIpcCloseMemHandle(r.peer_shards[p]);
      }
    }
    if (r.own_shard) (void)hipFree(r.own_shard);
    r.own_shard = nullptr;
    r.peer_shards.clear();
  }

  int rank_, world_, device_;
  std::unordered_map<int, IpcReg> regs_;
};

void register_ipc_ps(pybind11::module_& m) {
  namespace py = pybind11;
  py::class_<IpcPs>(m, "IpcPs")
      .def(py::init<int, int, int>(), py::arg("rank"), py::arg("world"),
           py::arg("device"))
      .def("register_local", &IpcPs::register_local)
      .def("open_peers", &IpcPs::open_peers,
           py::call_guard<py::gil_scoped_release>())
      .def("send", &IpcPs::send, py::call_guard<py::gil_scoped_release>())
      .def("receive", &IpcPs::receive, py::call_guard<py::gil_scoped_release>())
      .def("free_tensor", &IpcPs::free_tensor);
}

}  // namespace tma
