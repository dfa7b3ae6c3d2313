// RCCL-over-xGMI collective engine for MI355X.
//
// MI355X-native re-design of TorchMPI's GPU collective layer
// (lib/collectives_cuda.cpp — SURVEY.md sections 2.1, 2.4, 3.3). There is no
// CUDA-aware-MPI path, no staged-host path and no library-collective path:
// RCCL `ncclSend/ncclRecv` point-to-point over xGMI is the only transport,
// and every collective is composed here:
//   allreduce      = ring reduce-scatter + ring allgather, local reduction by
//                    our gfx950 kernel (kernels/reduce.hip), striped over R
//                    link-disjoint rings (xGMI is point-to-point: one ring is
//                    per-link bound at ~153 GB/s, so the headline path runs
//                    several rings on separate communicators/streams).
//   reduce_scatter = the first phase of the above.
//   allgather      = ring forwarding.
//   broadcast      = chunk-pipelined chain from the root.
//   reduce         = ring reduce-scatter + gather-to-root.
//   sendreceive    = grouped ncclSend+ncclRecv pair.
// Async semantics are pure stream semantics: every verb enqueues onto
// dedicated comm streams that first wait on an event recorded on the
// caller's current PyTorch stream, and returns a Work holding a done-event;
// Work.wait() makes the current stream wait (reference: the handle +
// syncHandle machinery of lib/collectives_cuda.cpp / thread_pool.h).
#include <torch/extension.h>

#include <ATen/ATen.h>
#include <ATen/hip/impl/HIPCachingAllocatorMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <cstdio>
#include <map>
#include <memory>
#include <unordered_map>
#include <vector>

#include "common.h"
#include "kernels.h"

namespace tma {

#define TMA_HIP_CHECK(cmd)                                                    \
  do {                                                                        \
    hipError_t e = (cmd);                                                     \
    TMA_CHECK(e == hipSuccess, std::string("HIP error: ") + hipGetErrorString(e)); \
  } while (0)

#define TMA_NCCL_CHECK(cmd)                                                   \
  do {                                                                        \
    ncclResult_t r = (cmd);                                                   \
    TMA_CHECK(r == ncclSuccess,                                               \
              std::string("RCCL error: ") + ncclGetErrorString(r));           \
  } while (0)

namespace {

DType dtype_of(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return DType::F32;
    case at::kDouble: return DType::F64;
    case at::kHalf: return DType::F16;
    case at::kBFloat16: return DType::BF16;
    case at::kInt: return DType::I32;
    case at::kLong: return DType::I64;
    case at::kByte: return DType::U8;
    case at::kChar: return DType::I8;
    default:
      throw std::runtime_error("torchmpi_amd: unsupported tensor dtype for collective");
  }
}

void check_gpu_tensor(const at::Tensor& t) {
  TMA_CHECK(t.is_cuda(), "expected a GPU tensor (CPU tensors use the TCP transport)");
  TMA_CHECK(t.is_contiguous(), "collectives require contiguous tensors");
}

// Split count elements into `parts` segments whose boundaries are multiples
// of `align` elements (except possibly the last).
void split_aligned(size_t count, int parts, size_t align, std::vector<size_t>& off,
                   std::vector<size_t>& cnt) {
  off.assign(parts, 0);
  cnt.assign(parts, 0);
  size_t per = (count + parts - 1) / parts;
  per = (per + align - 1) / align * align;
  size_t o = 0;
  for (int i = 0; i < parts; ++i) {
    off[i] = o;
    cnt[i] = o < count ? std::min(per, count - o) : 0;
    o += cnt[i];
  }
}

}  // namespace

// The single source of truth for the ring schedules (used by the engine AND
// exported to Python for simulation tests). phase 0 = reduce-scatter,
// phase 1 = allgather. Position p at step `step` sends segment send_seg to
// its ring successor and receives recv_seg from its predecessor; after the
// RS phase rank g owns segment g for any ring permutation.
inline void ring_step_segs(const std::vector<int>& perm, int W, int p, int step,
                           int phase, int& send_seg, int& recv_seg) {
  if (phase == 0) {
    send_seg = perm[(p - step - 1 + 2 * W) % W];
    recv_seg = perm[(p - step - 2 + 2 * W) % W];
  } else {
    send_seg = perm[(p - step + W) % W];
    recv_seg = perm[(p - step - 1 + 2 * W) % W];
  }
}

// Completion handle: one hipEvent per comm stream used by the op.
struct GpuWork {
  std::vector<hipEvent_t> events;
  std::vector<at::Tensor> keep;  // tensors the op reads/writes (+ scratch)
  bool synced = false;

  ~GpuWork() {
    for (auto ev : events) (void)hipEventDestroy(ev);
  }

  // Make the caller's current stream wait for completion (stream semantics —
  // the cheap, overlap-preserving sync used by the training loop).
  void wait() {
    auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA();
    for (auto ev : events) TMA_HIP_CHECK(hipStreamWaitEvent(cur.stream(), ev, 0));
    synced = true;
  }

  // Block the host until done (reference torchmpi_synchronize_handle's
  // strongest form).
  void wait_host() {
    for (auto ev : events) TMA_HIP_CHECK(hipEventSynchronize(ev));
    synced = true;
  }

  bool query() {
    for (auto ev : events) {
      hipError_t r = hipEventQuery(ev);
      if (r == hipErrorNotReady) return false;
      TMA_CHECK(r == hipSuccess, "hipEventQuery failed");
    }
    return true;
  }
};

class RcclEngine {
 public:
  // rings[r] is a permutation of [0, world) giving ring r's cycle order;
  // uids[r] is the ncclUniqueId for ring r's communicator.
  RcclEngine(int rank, int world, int device, const std::vector<std::string>& uids,
             const std::vector<std::vector<int>>& rings)
      : rank_(rank), world_(world), device_(device), rings_(rings) {
    TMA_HIP_CHECK(hipSetDevice(device_));
    TMA_CHECK(!uids.empty() && uids.size() == rings.size(),
              "need one uid per ring communicator");
    comms_.resize(uids.size());
    for (size_t r = 0; r < uids.size(); ++r) {
      TMA_CHECK(uids[r].size() == sizeof(ncclUniqueId), "bad nccl unique id size");
      ncclUniqueId id;
      std::memcpy(&id, uids[r].data(), sizeof(id));
      TMA_NCCL_CHECK(ncclCommInitRank(&comms_[r], world_, id, rank_));
      streams_.push_back(c10::hip::getStreamFromPoolMasqueradingAsCUDA(/*isHighPriority=*/true, device_));
      recv_streams_.push_back(
          c10::hip::getStreamFromPoolMasqueradingAsCUDA(true, device_));
      red_streams_.push_back(
          c10::hip::getStreamFromPoolMasqueradingAsCUDA(true, device_));
      // position of our rank in ring r, successor and predecessor
      const auto& perm = rings[r];
      TMA_CHECK(static_cast<int>(perm.size()) == world_, "ring size != world");
      int pos = -1;
      for (int i = 0; i < world_; ++i)
        if (perm[i] == rank_) pos = i;
      TMA_CHECK(pos >= 0, "rank missing from ring");
      pos_.push_back(pos);
      next_.push_back(perm[(pos + 1) % world_]);
      prev_.push_back(perm[(pos - 1 + world_) % world_]);
    }
    const char* cb = std::getenv("TORCHMPI_CHUNK_BYTES");
    chunk_bytes_ = cb ? static_cast<size_t>(std::atoll(cb)) : (4u << 20);
    const char* mr = std::getenv("TORCHMPI_MULTIRING_MIN_BYTES");
    multiring_min_bytes_ = mr ? static_cast<size_t>(std::atoll(mr)) : (1u << 20);
    const char* ge = std::getenv("TORCHMPI_GRAPH");
    graphs_enabled_ = !(ge && std::string(ge) == "0");
    const char* gm = std::getenv("TORCHMPI_GRAPH_MIN_BYTES");
    if (gm) graph_min_bytes_ = static_cast<size_t>(std::atoll(gm));
    const char* rh = std::getenv("TORCHMPI_RHD_MAX_BYTES");
    rhd_max_bytes_ = rh ? static_cast<size_t>(std::atoll(rh)) : (256u << 10);
    const char* ov = std::getenv("TORCHMPI_RING_OVERLAP");
    ring_overlap_ = ov && std::string(ov) == "1";
  }

  ~RcclEngine() {
    for (auto ev : fork_ev_)
      if (ev) (void)hipEventDestroy(ev);
    for (auto& v : recv_done_)
      for (auto ev : v) (void)hipEventDestroy(ev);
    for (auto& v : red_done_)
      for (auto ev : v) (void)hipEventDestroy(ev);
    if (order_ev_) (void)hipEventDestroy(order_ev_);
    if (join_ev_) (void)hipEventDestroy(join_ev_);
    for (auto& kv : graphs_) {
      if (kv.second.exec) (void)hipGraphExecDestroy(kv.second.exec);
      for (auto ev : kv.second.events) (void)hipEventDestroy(ev);
    }
    for (auto c : comms_) (void)ncclCommDestroy(c);
  }

  int rank() const { return rank_; }
  int size() const { return world_; }
  int num_rings() const { return static_cast<int>(comms_.size()); }
  bool graphs_enabled() const { return graphs_enabled_; }
  void set_graphs_enabled(bool v) { graphs_enabled_ = v; }
  // Cap the rings used per collective (bench-time fallback: a wrong
  // multi-ring result degrades to single-ring instead of no result).
  void set_max_rings(int v) { max_rings_ = v < 1 ? 1 : v; }

  // 1-GPU validation of the hipGraph path (tests/test_gpu_collectives.py):
  // captures {grouped self ncclSend/ncclRecv -> local reduce kernel} and
  // replays it `replays` times: t doubles per replay. Exercises RCCL-ops-
  // under-capture + instantiate + replay with the same machinery the
  // multi-ring allreduce uses. Returns false (and disables graphs) if
  // capture fails.
  bool graph_selfcheck(at::Tensor t, int replays) {
    check_gpu_tensor(t);
    const size_t bytes = static_cast<size_t>(t.numel()) * t.element_size();
    const DType d = dtype_of(t);
    at::Tensor scratch = at::empty_like(t);
    auto s0 = streams_[0];
    hipEvent_t ready = record_ready();
    TMA_HIP_CHECK(hipStreamWaitEvent(s0.stream(), ready, 0));
    TMA_HIP_CHECK(hipEventDestroy(ready));
    hipGraphExec_t exec = nullptr;
    bool capturing = false;
    try {
      TMA_HIP_CHECK(hipStreamBeginCapture(s0.stream(),
                                          hipStreamCaptureModeThreadLocal));
      capturing = true;
      TMA_NCCL_CHECK(ncclGroupStart());
      TMA_NCCL_CHECK(ncclSend(t.data_ptr(), bytes, ncclInt8, rank_, comms_[0],
                              s0.stream()));
      TMA_NCCL_CHECK(ncclRecv(scratch.data_ptr(), bytes, ncclInt8, rank_,
                              comms_[0], s0.stream()));
      TMA_NCCL_CHECK(ncclGroupEnd());
      launch_reduce(t.data_ptr(), scratch.data_ptr(), t.numel(), d, RedOp::SUM,
                    s0.stream());
      hipGraph_t g = nullptr;
      TMA_HIP_CHECK(hipStreamEndCapture(s0.stream(), &g));
      capturing = false;
      TMA_HIP_CHECK(hipGraphInstantiate(&exec, g, nullptr, nullptr, 0));
      TMA_HIP_CHECK(hipGraphDestroy(g));
      for (int i = 0; i < replays; ++i)
        TMA_HIP_CHECK(hipGraphLaunch(exec, s0.stream()));
      TMA_HIP_CHECK(hipStreamSynchronize(s0.stream()));
      (void)hipGraphExecDestroy(exec);
      return true;
    } catch (const std::exception&) {
      if (capturing) {
        hipGraph_t g = nullptr;
        (void)hipStreamEndCapture(s0.stream(), &g);
        if (g) (void)hipGraphDestroy(g);
      }
      if (exec) (void)hipGraphExecDestroy(exec);
      graphs_enabled_ = false;
      return false;
    }
  }

  // ---- verbs -------------------------------------------------------------

  std::shared_ptr<GpuWork> allreduce(at::Tensor t, RedOp op, double scale_alpha) {
    check_gpu_tensor(t);
    auto work = std::make_shared<GpuWork>();
    work->keep.push_back(t);
    const size_t count = static_cast<size_t>(t.numel());
    const DType d = dtype_of(t);
    const size_t es = dtype_size(d);

    if (world_ == 1 || count == 0) {
      if (scale_alpha != 1.0 && count > 0)
        launch_scale(t.data_ptr(), count, d, scale_alpha,
                     c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream());
      record_done_current(work);
      return work;
    }

    hipEvent_t ready = record_ready();

    // Small payloads: recursive halving-doubling (log2 W latency).
    if ((world_ & (world_ - 1)) == 0 && count * es <= rhd_max_bytes_ &&
        count >= static_cast<size_t>(world_)) {
      auto s = streams_[0];
      TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
      order_in(s);
      at::Tensor scratch = scratch_like(count * es, s);
      work->keep.push_back(scratch);
      rhd_allreduce(static_cast<char*>(t.data_ptr()), count, d, op,
                    static_cast<char*>(scratch.data_ptr()));
      if (scale_alpha != 1.0)
        launch_scale(t.data_ptr(), count, d, scale_alpha, s.stream());
      record_stream_uses(t, s);
      record_done(work, s);
      order_out(1);
      TMA_HIP_CHECK(hipEventDestroy(ready));
      return work;
    }

    const size_t align = 16 / es;  // keep 16B alignment for the reduce kernel
    int nrings = count * es >= multiring_min_bytes_ ? used_rings() : 1;

    std::vector<size_t> soff, scnt;
    split_aligned(count, nrings, align, soff, scnt);
    char* base = static_cast<char*>(t.data_ptr());

    // hipGraph replay path: the composed multi-ring schedule is ~20 launches
    // per ring; for stable buffers (gradient buckets, bench tensors) the
    // whole schedule is captured once and replayed with ONE launch
    // ("capture launch-bound inner loops in hipGraphs").
    if (graphs_enabled_ && count * es >= graph_min_bytes_) {
      GraphKey key{base, count, static_cast<int>(d), static_cast<int>(op),
                   scale_alpha, nrings};
      auto it = graphs_.find(key);
      if (it != graphs_.end()) {
        auto s0 = streams_[0];
        TMA_HIP_CHECK(hipStreamWaitEvent(s0.stream(), ready, 0));
        order_in(s0);
        TMA_HIP_CHECK(hipGraphLaunch(it->second.exec, s0.stream()));
        record_stream_uses(t, s0);
        work->keep.push_back(it->second.scratch);
        record_done(work, s0);
        order_out(1);
        TMA_HIP_CHECK(hipEventDestroy(ready));
        return work;
      }
    }

    for (int r = 0; r < nrings; ++r) {
      if (scnt[r] == 0) continue;
      auto s = streams_[r];
      TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
      order_in(s);
      ring_allreduce(r, base + soff[r] * es, scnt[r], d, op, work);
      if (scale_alpha != 1.0)
        launch_scale(base + soff[r] * es, scnt[r], d, scale_alpha, s.stream());
      record_stream_uses(t, s);
      record_done(work, s);
    }
    order_out(nrings);
    TMA_HIP_CHECK(hipEventDestroy(ready));

    // Capture for next time (after the eager run warmed all RCCL paths).
    if (graphs_enabled_ && count * es >= graph_min_bytes_ &&
        graphs_.size() < 1024) {
      try_capture_allreduce(base, count, d, op, scale_alpha, nrings, soff, scnt);
    }
    return work;
  }

  std::shared_ptr<GpuWork> reduce_scatter(at::Tensor input, at::Tensor output,
                                          RedOp op) {
    check_gpu_tensor(input);
    check_gpu_tensor(output);
    auto work = std::make_shared<GpuWork>();
    work->keep = {input, output};
    const size_t count = static_cast<size_t>(input.numel());
    const DType d = dtype_of(input);
    const size_t es = dtype_size(d);
    std::vector<size_t> off, cnt;
    split_segments(count, world_, off, cnt);
    TMA_CHECK(static_cast<size_t>(output.numel()) >= cnt[rank_],
              "reduce_scatter output too small");
    if (world_ == 1) {
      auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA();
      TMA_HIP_CHECK(hipMemcpyAsync(output.data_ptr(), input.data_ptr(), count * es,
                                   hipMemcpyDeviceToDevice, cur.stream()));
      record_done_current(work);
      return work;
    }
    hipEvent_t ready = record_ready();
    char* base = static_cast<char*>(input.data_ptr());
    const size_t align = 16 / es;
    int nrings = count * es >= multiring_min_bytes_ ? used_rings() : 1;
    // Ring r reduces part r of every global segment; the union of rank g's
    // parts is exactly its contiguous segment g.
    for (int r = 0; r < nrings; ++r) {
      std::vector<size_t> off_r, cnt_r;
      segment_parts(off, cnt, r, nrings, align, off_r, cnt_r);
      size_t maxc = 0;
      for (auto c : cnt_r) maxc = std::max(maxc, c);
      if (maxc == 0) continue;
      auto s = streams_[r];
      TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
      order_in(s);
      at::Tensor scratch = scratch_like(2 * maxc * es, s);
      work->keep.push_back(scratch);
      if (ring_overlap_ && world_ > 2)
        ring_rs_overlap(r, base, off_r, cnt_r, d, op,
                        static_cast<char*>(scratch.data_ptr()));
      else
        ring_rs(r, base, off_r, cnt_r, d, op,
                static_cast<char*>(scratch.data_ptr()));
      record_stream_uses(input, s);
      record_done(work, s);
    }
    // Copy the (now fully reduced) contiguous own segment into the output,
    // after all rings finish: join rings on stream 0 via the work events.
    auto s0 = streams_[0];
    for (auto ev : work->events)
      TMA_HIP_CHECK(hipStreamWaitEvent(s0.stream(), ev, 0));
    TMA_HIP_CHECK(hipMemcpyAsync(output.data_ptr(), base + off[rank_] * es,
                                 cnt[rank_] * es, hipMemcpyDeviceToDevice,
                                 s0.stream()));
    record_stream_uses(output, s0);
    record_done(work, s0);
    order_out(1);
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

  std::shared_ptr<GpuWork> broadcast(at::Tensor t, int root) {
    check_gpu_tensor(t);
    auto work = std::make_shared<GpuWork>();
    work->keep.push_back(t);
    const size_t bytes = static_cast<size_t>(t.numel()) * t.element_size();
    if (world_ == 1 || bytes == 0) {
      record_done_current(work);
      return work;
    }
    hipEvent_t ready = record_ready();
    // Multi-chain: stripe the tensor over the link-disjoint rings; each
    // stripe flows down ring r''s chain (root first) chunk-pipelined.
    char* base = static_cast<char*>(t.data_ptr());
    int nrings = bytes >= multiring_min_bytes_ ? used_rings() : 1;
    std::vector<size_t> soff, scnt;
    split_aligned(bytes, nrings, 16, soff, scnt);
    for (int r = 0; r < nrings; ++r) {
      if (scnt[r] == 0) continue;
      auto s = streams_[r];
      TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
      order_in(s);
      chain_broadcast(r, base + soff[r], scnt[r], root);
      record_stream_uses(t, s);
      record_done(work, s);
    }
    order_out(nrings);
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

  // Chunk-pipelined chain broadcast of `bytes` at `base` over ring r.
  void chain_broadcast(int r, char* base, size_t bytes, int root) {
    auto s = streams_[r];
    const auto& perm = rings_[r];
    int rootpos = 0;
    for (int i = 0; i < world_; ++i)
      if (perm[i] == root) rootpos = i;
    int mypos = (pos_[r] - rootpos + world_) % world_;
    int nxt = next_[r], prv = prev_[r];
    size_t nchunks = (bytes + chunk_bytes_ - 1) / chunk_bytes_;
    for (size_t c = 0; c <= nchunks; ++c) {
      bool do_recv = mypos > 0 && c < nchunks;
      bool do_send = mypos < world_ - 1 && c > 0;
      if (!do_recv && !do_send) continue;
      TMA_NCCL_CHECK(ncclGroupStart());
      if (do_recv) {
        size_t o = c * chunk_bytes_;
        size_t n = std::min(chunk_bytes_, bytes - o);
        TMA_NCCL_CHECK(ncclRecv(base + o, n, ncclInt8, prv, comms_[r], s.stream()));
      }
      if (do_send) {
        size_t o = (c - 1) * chunk_bytes_;
        size_t n = std::min(chunk_bytes_, bytes - o);
        TMA_NCCL_CHECK(ncclSend(base + o, n, ncclInt8, nxt, comms_[r], s.stream()));
      }
      TMA_NCCL_CHECK(ncclGroupEnd());
    }
  }

  std::shared_ptr<GpuWork> reduce(at::Tensor t, RedOp op, int root) {
    check_gpu_tensor(t);
    auto work = std::make_shared<GpuWork>();
    work->keep.push_back(t);
    const size_t count = static_cast<size_t>(t.numel());
    const DType d = dtype_of(t);
    const size_t es = dtype_size(d);
    if (world_ == 1 || count == 0) {
      record_done_current(work);
      return work;
    }
    hipEvent_t ready = record_ready();
    auto s0 = streams_[0];
    TMA_HIP_CHECK(hipStreamWaitEvent(s0.stream(), ready, 0));
    order_in(s0);

    // Non-root ranks work on a scratch copy so their input is preserved
    // (MPI_Reduce semantics, reference lib/collectives.cpp reduce). The
    // root reduces in place: its tensor becomes the result anyway.
    char* tbase = static_cast<char*>(t.data_ptr());
    char* base = tbase;
    const size_t align = 16 / es;
    int nrings = count * es >= multiring_min_bytes_ ? used_rings() : 1;
    at::Tensor scratch;
    if (rank_ != root) {
      scratch = scratch_like(count * es, s0);
      // The scratch copy is READ by every ring stream below — record each
      // so the caching allocator cannot reuse it early.
      for (int r = 1; r < nrings; ++r)
        record_stream_uses(scratch, streams_[r]);
      work->keep.push_back(scratch);
      TMA_HIP_CHECK(hipMemcpyAsync(scratch.data_ptr(), tbase, count * es,
                                   hipMemcpyDeviceToDevice, s0.stream()));
      base = static_cast<char*>(scratch.data_ptr());
    }
    record_stream_uses(t, s0);  // s0's memcpy/recv reads t

    // Multi-ring (same striping as allreduce): ring r reduce-scatters its
    // stripe then gathers the stripe's segments to root over comms_[r],
    // using all xGMI links instead of one.
    std::vector<size_t> soff, scnt;
    split_aligned(count, nrings, align, soff, scnt);
    hipEvent_t fork = nullptr;
    if (nrings > 1) {
      TMA_HIP_CHECK(hipEventCreateWithFlags(&fork, hipEventDisableTiming));
      TMA_HIP_CHECK(hipEventRecord(fork, s0.stream()));
    }
    for (int r = 0; r < nrings; ++r) {
      if (scnt[r] == 0) continue;
      auto s = streams_[r];
      if (r > 0) {
        TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), fork, 0));
        order_in(s);
      }
      ring_reduce_scatter(r, base + soff[r] * es, scnt[r], d, op, work);
      // Gather stripe segments to root: after RS, rank g owns segment g.
      std::vector<size_t> off, cnt;
      split_segments(scnt[r], world_, off, cnt);
      TMA_NCCL_CHECK(ncclGroupStart());
      if (rank_ == root) {
        for (int g = 0; g < world_; ++g) {
          if (g == root || cnt[g] == 0) continue;
          TMA_NCCL_CHECK(ncclRecv(tbase + (soff[r] + off[g]) * es, cnt[g] * es,
                                  ncclInt8, g, comms_[r], s.stream()));
        }
      } else {
        if (cnt[rank_] > 0)
          TMA_NCCL_CHECK(ncclSend(base + (soff[r] + off[rank_]) * es,
                                  cnt[rank_] * es, ncclInt8, root, comms_[r],
                                  s.stream()));
      }
      TMA_NCCL_CHECK(ncclGroupEnd());
      record_stream_uses(t, s);
      record_done(work, s);
    }
    order_out(nrings);
    if (fork) TMA_HIP_CHECK(hipEventDestroy(fork));
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

  std::shared_ptr<GpuWork> allgather(at::Tensor input, at::Tensor output) {
    check_gpu_tensor(input);
    check_gpu_tensor(output);
    auto work = std::make_shared<GpuWork>();
    work->keep = {input, output};
    const size_t bytes = static_cast<size_t>(input.numel()) * input.element_size();
    TMA_CHECK(static_cast<size_t>(output.numel()) * output.element_size() ==
                  bytes * static_cast<size_t>(world_),
              "allgather output must be world_size x input");
    auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA();
    char* ob = static_cast<char*>(output.data_ptr());
    TMA_HIP_CHECK(hipMemcpyAsync(ob + static_cast<size_t>(rank_) * bytes,
                                 input.data_ptr(), bytes, hipMemcpyDeviceToDevice,
                                 cur.stream()));
    if (world_ == 1) {
      record_done_current(work);
      return work;
    }
    hipEvent_t ready = record_ready();
    // Multi-ring: slot g is a "global segment" owned by rank g; ring r
    // forwards part r of every slot (ring_ag), using all xGMI links.
    std::vector<size_t> off(world_), cnt(world_);
    size_t elems = static_cast<size_t>(input.numel());
    size_t es2 = static_cast<size_t>(input.element_size());
    for (int g = 0; g < world_; ++g) {
      off[g] = static_cast<size_t>(g) * elems;
      cnt[g] = elems;
    }
    const size_t align = 16 / es2 ? 16 / es2 : 1;
    int nrings = bytes >= multiring_min_bytes_ ? used_rings() : 1;
    for (int r = 0; r < nrings; ++r) {
      std::vector<size_t> off_r, cnt_r;
      segment_parts(off, cnt, r, nrings, align, off_r, cnt_r);
      size_t maxc = 0;
      for (auto c : cnt_r) maxc = std::max(maxc, c);
      if (maxc == 0) continue;
      auto s = streams_[r];
      TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
      order_in(s);
      ring_ag(r, ob, off_r, cnt_r, es2);
      record_stream_uses(output, s);
      record_done(work, s);
    }
    order_out(nrings);
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

  std::shared_ptr<GpuWork> sendrecv(at::Tensor send_t, int dst, at::Tensor recv_t,
                                    int src) {
    check_gpu_tensor(send_t);
    check_gpu_tensor(recv_t);
    auto work = std::make_shared<GpuWork>();
    work->keep = {send_t, recv_t};
    hipEvent_t ready = record_ready();
    auto s = streams_[0];
    TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
    order_in(s);
    size_t sbytes = static_cast<size_t>(send_t.numel()) * send_t.element_size();
    size_t rbytes = static_cast<size_t>(recv_t.numel()) * recv_t.element_size();
    TMA_NCCL_CHECK(ncclGroupStart());
    TMA_NCCL_CHECK(
        ncclSend(send_t.data_ptr(), sbytes, ncclInt8, dst, comms_[0], s.stream()));
    TMA_NCCL_CHECK(
        ncclRecv(recv_t.data_ptr(), rbytes, ncclInt8, src, comms_[0], s.stream()));
    TMA_NCCL_CHECK(ncclGroupEnd());
    record_stream_uses(send_t, s);
    record_stream_uses(recv_t, s);
    record_done(work, s);
    order_out(1);
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

  // Gather to root / scatter from root / all-to-all: grouped p2p fans
  // (reference MPI_Gather/Scatter/Alltoall verb parity; xGMI is fully
  // connected so every pair is one direct link).
  std::shared_ptr<GpuWork> gather(at::Tensor input, at::Tensor output, int root) {
    check_gpu_tensor(input);
    auto work = std::make_shared<GpuWork>();
    work->keep = {input, output};
    size_t bytes = static_cast<size_t>(input.numel()) * input.element_size();
    hipEvent_t ready = record_ready();
    auto s = streams_[0];
    TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
    order_in(s);
    TMA_NCCL_CHECK(ncclGroupStart());
    if (rank_ == root) {
      check_gpu_tensor(output);
      char* ob = static_cast<char*>(output.data_ptr());
      TMA_HIP_CHECK(hipMemcpyAsync(ob + static_cast<size_t>(root) * bytes,
                                   input.data_ptr(), bytes,
                                   hipMemcpyDeviceToDevice, s.stream()));
      for (int r = 0; r < world_; ++r)
        if (r != root)
          TMA_NCCL_CHECK(ncclRecv(ob + static_cast<size_t>(r) * bytes, bytes,
                                  ncclInt8, r, comms_[0], s.stream()));
      record_stream_uses(output, s);
    } else {
      TMA_NCCL_CHECK(
          ncclSend(input.data_ptr(), bytes, ncclInt8, root, comms_[0], s.stream()));
    }
    TMA_NCCL_CHECK(ncclGroupEnd());
    record_stream_uses(input, s);
    record_done(work, s);
    order_out(1);
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

  std::shared_ptr<GpuWork> scatter(at::Tensor input, at::Tensor output, int root) {
    check_gpu_tensor(output);
    auto work = std::make_shared<GpuWork>();
    work->keep = {input, output};
    size_t bytes = static_cast<size_t>(output.numel()) * output.element_size();
    hipEvent_t ready = record_ready();
    auto s = streams_[0];
    TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
    order_in(s);
    TMA_NCCL_CHECK(ncclGroupStart());
    if (rank_ == root) {
      check_gpu_tensor(input);
      char* ib = static_cast<char*>(input.data_ptr());
      TMA_HIP_CHECK(hipMemcpyAsync(output.data_ptr(),
                                   ib + static_cast<size_t>(root) * bytes, bytes,
                                   hipMemcpyDeviceToDevice, s.stream()));
      for (int r = 0; r < world_; ++r)
        if (r != root)
          TMA_NCCL_CHECK(ncclSend(ib + static_cast<size_t>(r) * bytes, bytes,
                                  ncclInt8, r, comms_[0], s.stream()));
      record_stream_uses(input, s);
    } else {
      TMA_NCCL_CHECK(ncclRecv(output.data_ptr(), bytes, ncclInt8, root, comms_[0],
                              s.stream()));
    }
    TMA_NCCL_CHECK(ncclGroupEnd());
    record_stream_uses(output, s);
    record_done(work, s);
    order_out(1);
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

  std::shared_ptr<GpuWork> alltoall(at::Tensor input, at::Tensor output) {
    check_gpu_tensor(input);
    check_gpu_tensor(output);
    auto work = std::make_shared<GpuWork>();
    work->keep = {input, output};
    TMA_CHECK(input.numel() % world_ == 0 && output.numel() == input.numel(),
              "alltoall: numel must be divisible by world_size and match");
    size_t bytes =
        static_cast<size_t>(input.numel()) * input.element_size() / world_;
    hipEvent_t ready = record_ready();
    auto s = streams_[0];
    TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
    order_in(s);
    const char* ib = static_cast<const char*>(input.data_ptr());
    char* ob = static_cast<char*>(output.data_ptr());
    TMA_HIP_CHECK(hipMemcpyAsync(ob + static_cast<size_t>(rank_) * bytes,
                                 ib + static_cast<size_t>(rank_) * bytes, bytes,
                                 hipMemcpyDeviceToDevice, s.stream()));
    TMA_NCCL_CHECK(ncclGroupStart());
    for (int r = 0; r < world_; ++r) {
      if (r == rank_) continue;
      TMA_NCCL_CHECK(ncclSend(ib + static_cast<size_t>(r) * bytes, bytes, ncclInt8,
                              r, comms_[0], s.stream()));
      TMA_NCCL_CHECK(ncclRecv(ob + static_cast<size_t>(r) * bytes, bytes, ncclInt8,
                              r, comms_[0], s.stream()));
    }
    TMA_NCCL_CHECK(ncclGroupEnd());
    record_stream_uses(input, s);
    record_stream_uses(output, s);
    record_done(work, s);
    order_out(1);
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

  std::shared_ptr<GpuWork> send(at::Tensor t, int dst) {
    check_gpu_tensor(t);
    auto work = std::make_shared<GpuWork>();
    work->keep = {t};
    hipEvent_t ready = record_ready();
    auto s = streams_[0];
    TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
    order_in(s);
    TMA_NCCL_CHECK(ncclSend(t.data_ptr(),
                            static_cast<size_t>(t.numel()) * t.element_size(), ncclInt8,
                            dst, comms_[0], s.stream()));
    record_stream_uses(t, s);
    record_done(work, s);
    order_out(1);
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

  std::shared_ptr<GpuWork> recv(at::Tensor t, int src) {
    check_gpu_tensor(t);
    auto work = std::make_shared<GpuWork>();
    work->keep = {t};
    hipEvent_t ready = record_ready();
    auto s = streams_[0];
    TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), ready, 0));
    order_in(s);
    TMA_NCCL_CHECK(ncclRecv(t.data_ptr(),
                            static_cast<size_t>(t.numel()) * t.element_size(), ncclInt8,
                            src, comms_[0], s.stream()));
    record_stream_uses(t, s);
    record_done(work, s);
    order_out(1);
    TMA_HIP_CHECK(hipEventDestroy(ready));
    return work;
  }

 private:
  // Record an event on the caller's current stream (orders the collective
  // after e.g. the backward kernels that produced the gradient — reference
  // stack 3.3's ev_ready event record on currentStream).
  hipEvent_t record_ready() {
    hipEvent_t ev;
    TMA_HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    TMA_HIP_CHECK(hipEventRecord(ev, c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream()));
    return ev;
  }

  void record_done(std::shared_ptr<GpuWork>& w, c10::hip::HIPStreamMasqueradingAsCUDA s) {
    hipEvent_t ev;
    TMA_HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    TMA_HIP_CHECK(hipEventRecord(ev, s.stream()));
    w->events.push_back(ev);
  }

  void record_done_current(std::shared_ptr<GpuWork>& w) {
    hipEvent_t ev;
    TMA_HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    TMA_HIP_CHECK(hipEventRecord(ev, c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream()));
    w->events.push_back(ev);
  }

  // ---- inter-collective ordering -----------------------------------------
  // Collectives on this engine must execute in issue order on every rank
  // (grouped-p2p matching is per-communicator call order). Per-stream FIFO
  // gives that while everything is eager, but hipGraph replays root/join
  // only on streams_[0]; order_in/order_out chain every verb through one
  // persistent event so replays and eager multi-stream ops never overlap on
  // a communicator.
  void order_in(c10::hip::HIPStreamMasqueradingAsCUDA s) {
    if (order_ev_) TMA_HIP_CHECK(hipStreamWaitEvent(s.stream(), order_ev_, 0));
  }

  // Join streams [1, nrings) into streams_[0] and mark the op boundary.
  void order_out(int nrings_used) {
    if (!order_ev_)
      TMA_HIP_CHECK(hipEventCreateWithFlags(&order_ev_, hipEventDisableTiming));
    auto s0 = streams_[0];
    for (int r = 1; r < nrings_used; ++r) {
      if (!join_ev_)
        TMA_HIP_CHECK(hipEventCreateWithFlags(&join_ev_, hipEventDisableTiming));
      TMA_HIP_CHECK(hipEventRecord(join_ev_, streams_[r].stream()));
      TMA_HIP_CHECK(hipStreamWaitEvent(s0.stream(), join_ev_, 0));
    }
    TMA_HIP_CHECK(hipEventRecord(order_ev_, s0.stream()));
  }

  void record_stream_uses(const at::Tensor& t, c10::hip::HIPStreamMasqueradingAsCUDA s) {
    c10::hip::HIPCachingAllocatorMasqueradingAsCUDA::recordStreamMasqueradingAsCUDA(
        t.storage().data_ptr(), s);
  }

  at::Tensor scratch_like(size_t bytes, c10::hip::HIPStreamMasqueradingAsCUDA s) {
    auto t = at::empty({static_cast<int64_t>(bytes)},
                       at::TensorOptions().dtype(at::kByte).device(at::kCUDA, device_));
    record_stream_uses(t, s);
    return t;
  }

  // Per-ring segment tables for multi-ring verbs: the W global segments
  // (indexed by rank) are each split into `nrings` aligned parts; ring r
  // moves part r of every segment. Parts of one segment are contiguous, so
  // the union over rings of rank g's parts is exactly global segment g.
  void segment_parts(const std::vector<size_t>& off, const std::vector<size_t>& cnt,
                     int r, int nrings, size_t align, std::vector<size_t>& off_r,
                     std::vector<size_t>& cnt_r) {
    const int W = static_cast<int>(off.size());
    off_r.resize(W);
    cnt_r.resize(W);
    std::vector<size_t> po, pc;
    for (int g = 0; g < W; ++g) {
      split_aligned(cnt[g], nrings, align, po, pc);
      off_r[g] = off[g] + po[r];
      cnt_r[g] = pc[r];
    }
  }

  // Ring reduce-scatter on ring r over the segment table (off/cnt indexed by
  // GLOBAL RANK, elements). The schedule walks segments in ring order (at
  // step s, position p sends segment perm[(p-s-1) mod W]), so after W-1
  // steps rank g owns the fully reduced segment g — on ANY permutation.
  // `sc` is a double-buffer scratch of >= 2*max(cnt)*es bytes.
  void ring_rs(int r, char* base, const std::vector<size_t>& off,
               const std::vector<size_t>& cnt, DType d, RedOp op, char* sc) {
    const size_t es = dtype_size(d);
    const int W = world_;
    auto s = streams_[r];
    const auto& perm = rings_[r];
    size_t maxc = 0;
    for (auto c : cnt) maxc = std::max(maxc, c);
    const int p = pos_[r];
    for (int step = 0; step < W - 1; ++step) {
      int send_seg, recv_seg;
      ring_step_segs(perm, W, p, step, /*phase=*/0, send_seg, recv_seg);
      char* rbuf = sc + (step % 2) * maxc * es;
      TMA_NCCL_CHECK(ncclGroupStart());
      if (cnt[send_seg] > 0)
        TMA_NCCL_CHECK(ncclSend(base + off[send_seg] * es, cnt[send_seg] * es,
                                ncclInt8, next_[r], comms_[r], s.stream()));
      if (cnt[recv_seg] > 0)
        TMA_NCCL_CHECK(ncclRecv(rbuf, cnt[recv_seg] * es, ncclInt8, prev_[r],
                                comms_[r], s.stream()));
      TMA_NCCL_CHECK(ncclGroupEnd());
      if (cnt[recv_seg] > 0)
        launch_reduce(base + off[recv_seg] * es, rbuf, cnt[recv_seg], d, op,
                      s.stream());
    }
  }

  // Ring allgather on ring r over the segment table: rank g starts owning
  // segment g; forwards owned segments around the ring.
  void ring_ag(int r, char* base, const std::vector<size_t>& off,
               const std::vector<size_t>& cnt, size_t es) {
    const int W = world_;
    auto s = streams_[r];
    const auto& perm = rings_[r];
    const int p = pos_[r];
    for (int step = 0; step < W - 1; ++step) {
      int send_seg, recv_seg;
      ring_step_segs(perm, W, p, step, /*phase=*/1, send_seg, recv_seg);
      TMA_NCCL_CHECK(ncclGroupStart());
      if (cnt[send_seg] > 0)
        TMA_NCCL_CHECK(ncclSend(base + off[send_seg] * es, cnt[send_seg] * es,
                                ncclInt8, next_[r], comms_[r], s.stream()));
      if (cnt[recv_seg] > 0)
        TMA_NCCL_CHECK(ncclRecv(base + off[recv_seg] * es, cnt[recv_seg] * es,
                                ncclInt8, prev_[r], comms_[r], s.stream()));
      TMA_NCCL_CHECK(ncclGroupEnd());
    }
  }

  // Overlapped ring reduce-scatter (TORCHMPI_RING_OVERLAP=1, experimental —
  // flip on after measuring on the 8-GPU node): sends run on streams_[r],
  // recvs on recv_streams_[r], local reductions on red_streams_[r], so the
  // step-s reduction overlaps the step-s+1 transfer. Dependencies:
  //   send(s)  waits reduce(s-1)   (it sends the segment reduced last step)
  //   recv(s)  waits reduce(s-2)   (double-buffer slot reuse)
  //   reduce(s) waits recv(s)
  // Grouped-p2p matching holds: every rank issues send-group then recv-group
  // per step, so per-peer per-direction call order is identical everywhere.
  void ring_rs_overlap(int r, char* base, const std::vector<size_t>& off,
                       const std::vector<size_t>& cnt, DType d, RedOp op,
                       char* sc) {
    const size_t es = dtype_size(d);
    const int W = world_;
    auto sA = streams_[r];
    auto sB = recv_streams_[r];
    auto sR = red_streams_[r];
    const auto& perm = rings_[r];
    size_t maxc = 0;
    for (auto c : cnt) maxc = std::max(maxc, c);
    const int p = pos_[r];
    ensure_overlap_events(r, W - 1);
    auto& er = recv_done_[r];
    auto& dr = red_done_[r];
    hipEvent_t e_in = fork_ev_[r];
    TMA_HIP_CHECK(hipEventRecord(e_in, sA.stream()));
    TMA_HIP_CHECK(hipStreamWaitEvent(sB.stream(), e_in, 0));
    TMA_HIP_CHECK(hipStreamWaitEvent(sR.stream(), e_in, 0));
    for (int step = 0; step < W - 1; ++step) {
      int send_seg, recv_seg;
      ring_step_segs(perm, W, p, step, /*phase=*/0, send_seg, recv_seg);
      char* rbuf = sc + (step % 2) * maxc * es;
      if (step > 0)
        TMA_HIP_CHECK(hipStreamWaitEvent(sA.stream(), dr[step - 1], 0));
      if (cnt[send_seg] > 0) {
        TMA_NCCL_CHECK(ncclGroupStart());
        TMA_NCCL_CHECK(ncclSend(base + off[send_seg] * es, cnt[send_seg] * es,
                                ncclInt8, next_[r], comms_[r], sA.stream()));
        TMA_NCCL_CHECK(ncclGroupEnd());
      }
      if (step >= 2)
        TMA_HIP_CHECK(hipStreamWaitEvent(sB.stream(), dr[step - 2], 0));
      if (cnt[recv_seg] > 0) {
        TMA_NCCL_CHECK(ncclGroupStart());
        TMA_NCCL_CHECK(ncclRecv(rbuf, cnt[recv_seg] * es, ncclInt8, prev_[r],
                                comms_[r], sB.stream()));
        TMA_NCCL_CHECK(ncclGroupEnd());
      }
      TMA_HIP_CHECK(hipEventRecord(er[step], sB.stream()));
      TMA_HIP_CHECK(hipStreamWaitEvent(sR.stream(), er[step], 0));
      if (cnt[recv_seg] > 0)
        launch_reduce(base + off[recv_seg] * es, rbuf, cnt[recv_seg], d, op,
                      sR.stream());
      TMA_HIP_CHECK(hipEventRecord(dr[step], sR.stream()));
    }
    // Join the aux streams back into the ring's main stream (the allgather
    // phase and order_out continue on streams_[r]).
    TMA_HIP_CHECK(hipStreamWaitEvent(sA.stream(), dr[W - 2], 0));
  }

  void ensure_overlap_events(int r, int steps) {
    if ((int)fork_ev_.size() <= r) {
      fork_ev_.resize(num_rings(), nullptr);
      recv_done_.resize(num_rings());
      red_done_.resize(num_rings());
    }
    if (!fork_ev_[r])
      TMA_HIP_CHECK(hipEventCreateWithFlags(&fork_ev_[r], hipEventDisableTiming));
    while ((int)recv_done_[r].size() < steps) {
      hipEvent_t a, b;
      TMA_HIP_CHECK(hipEventCreateWithFlags(&a, hipEventDisableTiming));
      TMA_HIP_CHECK(hipEventCreateWithFlags(&b, hipEventDisableTiming));
      recv_done_[r].push_back(a);
      red_done_[r].push_back(b);
    }
  }

  void ring_reduce_scatter_buf(int r, char* base, size_t count, DType d, RedOp op,
                               char* sc) {
    std::vector<size_t> off, cnt;
    split_segments(count, world_, off, cnt);
    if (ring_overlap_ && world_ > 2)
      ring_rs_overlap(r, base, off, cnt, d, op, sc);
    else
      ring_rs(r, base, off, cnt, d, op, sc);
  }

  // Recursive halving-doubling allreduce on comms_[0]/streams_[0]:
  // log2(W) exchange steps (vs the ring's 2(W-1)) — the latency algorithm
  // for small payloads (north star: "ring and recursive-halving allreduce").
  // Same schedule as CpuComm::allreduce_rhd (tests cover it at W=2/4/8 over
  // the TCP transport). Requires W a power of two; `sc` >= count*es bytes.
  void rhd_allreduce(char* base, size_t count, DType d, RedOp op, char* sc) {
    const size_t es = dtype_size(d);
    const int W = world_;
    auto s = streams_[0];
    std::vector<size_t> off, cnt;
    split_segments(count, W, off, cnt);
    auto range_bytes = [&](int a, int b, size_t& o, size_t& n) {
      o = off[a] * es;
      n = (off[b - 1] + cnt[b - 1] - off[a]) * es;
    };
    int logw = 0;
    while ((1 << logw) < W) ++logw;
    int s0 = 0, s1 = W;
    for (int k = 0; k < logw; ++k) {
      int partner = rank_ ^ (1 << k);
      int mid = (s0 + s1) / 2;
      int keep0, keep1, send0, send1;
      if (((rank_ >> k) & 1) == 0) {
        keep0 = s0; keep1 = mid; send0 = mid; send1 = s1;
      } else {
        keep0 = mid; keep1 = s1; send0 = s0; send1 = mid;
      }
      size_t so, sn, ko, kn;
      range_bytes(send0, send1, so, sn);
      range_bytes(keep0, keep1, ko, kn);
      TMA_NCCL_CHECK(ncclGroupStart());
      if (sn > 0)
        TMA_NCCL_CHECK(ncclSend(base + so, sn, ncclInt8, partner, comms_[0],
                                s.stream()));
      if (kn > 0)
        TMA_NCCL_CHECK(ncclRecv(sc, kn, ncclInt8, partner, comms_[0],
                                s.stream()));
      TMA_NCCL_CHECK(ncclGroupEnd());
      if (kn > 0)
        launch_reduce(base + ko, sc, kn / es, d, op, s.stream());
      s0 = keep0;
      s1 = keep1;
    }
    for (int k = logw - 1; k >= 0; --k) {
      int partner = rank_ ^ (1 << k);
      int width = s1 - s0;
      int sib0 = (s0 % (2 * width) == 0) ? s0 + width : s0 - width;
      int sib1 = sib0 + width;
      size_t mo, mn, po, pn;
      range_bytes(s0, s1, mo, mn);
      range_bytes(sib0, sib1, po, pn);
      TMA_NCCL_CHECK(ncclGroupStart());
      if (mn > 0)
        TMA_NCCL_CHECK(ncclSend(base + mo, mn, ncclInt8, partner, comms_[0],
                                s.stream()));
      if (pn > 0)
        TMA_NCCL_CHECK(ncclRecv(base + po, pn, ncclInt8, partner, comms_[0],
                                s.stream()));
      TMA_NCCL_CHECK(ncclGroupEnd());
      s0 = std::min(s0, sib0);
      s1 = std::max(s1, sib1);
    }
  }

  // Eager wrappers: allocate per-call scratch from the caching allocator.
  void ring_reduce_scatter(int r, char* base, size_t count, DType d, RedOp op,
                           std::shared_ptr<GpuWork>& work) {
    std::vector<size_t> off, cnt;
    split_segments(count, world_, off, cnt);
    at::Tensor scratch = scratch_like(2 * cnt[0] * dtype_size(d), streams_[r]);
    work->keep.push_back(scratch);
    ring_reduce_scatter_buf(r, base, count, d, op,
                            static_cast<char*>(scratch.data_ptr()));
  }

  void ring_allreduce(int r, char* base, size_t count, DType d, RedOp op,
                      std::shared_ptr<GpuWork>& work) {
    std::vector<size_t> off, cnt;
    split_segments(count, world_, off, cnt);
    at::Tensor scratch = scratch_like(2 * cnt[0] * dtype_size(d), streams_[r]);
    work->keep.push_back(scratch);
    ring_allreduce_buf(r, base, count, d, op,
                       static_cast<char*>(scratch.data_ptr()));
  }

  // Full ring allreduce on ring r (reduce-scatter + allgather).
  void ring_allreduce_buf(int r, char* base, size_t count, DType d, RedOp op,
                          char* sc) {
    std::vector<size_t> off, cnt;
    split_segments(count, world_, off, cnt);
    if (ring_overlap_ && world_ > 2)
      ring_rs_overlap(r, base, off, cnt, d, op, sc);
    else
      ring_rs(r, base, off, cnt, d, op, sc);
    ring_ag(r, base, off, cnt, dtype_size(d));
  }

  struct GraphKey {
    void* ptr;
    size_t count;
    int dtype, op;
    double alpha;
    int nrings;
    bool operator==(const GraphKey& o) const {
      return ptr == o.ptr && count == o.count && dtype == o.dtype &&
             op == o.op && alpha == o.alpha && nrings == o.nrings;
    }
  };
  struct GraphKeyHash {
    size_t operator()(const GraphKey& k) const {
      size_t h = std::hash<void*>()(k.ptr);
      h = h * 31 + std::hash<size_t>()(k.count);
      h = h * 31 + static_cast<size_t>(k.dtype * 7 + k.op * 131 + k.nrings);
      h = h * 31 + std::hash<double>()(k.alpha);
      return h;
    }
  };
  struct GraphEntry {
    hipGraphExec_t exec = nullptr;
    at::Tensor scratch;
    std::vector<hipEvent_t> events;
  };

  // Capture the full multi-ring schedule into a hipGraph; on any failure the
  // graph path is disabled for the rest of the process (eager stays correct).
  void try_capture_allreduce(char* base, size_t count, DType d, RedOp op,
                             double alpha, int nrings,
                             const std::vector<size_t>& soff,
                             const std::vector<size_t>& scnt) {
    const size_t es = dtype_size(d);
    GraphEntry entry;
    // Graph-captured pointers must stay valid forever, but replays of
    // different graphs serialize on the same comm streams, so entries can
    // SHARE scratch: one persistent buffer per power-of-2 size class.
    std::vector<size_t> off0, cnt0;
    split_segments(scnt[0], world_, off0, cnt0);
    size_t slice = 2 * cnt0[0] * es + 256;
    size_t need = slice * nrings;
    size_t cls = 4096;
    while (cls < need) cls <<= 1;
    auto pit = graph_scratch_pool_.find(cls);
    if (pit == graph_scratch_pool_.end()) {
      at::Tensor scr = at::empty(
          {static_cast<int64_t>(cls)},
          at::TensorOptions().dtype(at::kByte).device(at::kCUDA, device_));
      pit = graph_scratch_pool_.emplace(cls, scr).first;
    }
    entry.scratch = pit->second;
    auto s0 = streams_[0];
    bool capturing = false;
    try {
      TMA_HIP_CHECK(hipStreamBeginCapture(s0.stream(),
                                          hipStreamCaptureModeThreadLocal));
      capturing = true;
      hipEvent_t e0;
      TMA_HIP_CHECK(hipEventCreateWithFlags(&e0, hipEventDisableTiming));
      entry.events.push_back(e0);
      TMA_HIP_CHECK(hipEventRecord(e0, s0.stream()));
      for (int r = 1; r < nrings; ++r)
        TMA_HIP_CHECK(hipStreamWaitEvent(streams_[r].stream(), e0, 0));
      char* sc = static_cast<char*>(entry.scratch.data_ptr());
      for (int r = 0; r < nrings; ++r) {
        if (scnt[r] == 0) continue;
        ring_allreduce_buf(r, base + soff[r] * es, scnt[r], d, op,
                           sc + static_cast<size_t>(r) * slice);
        if (alpha != 1.0)
          launch_scale(base + soff[r] * es, scnt[r], d, alpha,
                       streams_[r].stream());
      }
      for (int r = 1; r < nrings; ++r) {
        hipEvent_t er;
        TMA_HIP_CHECK(hipEventCreateWithFlags(&er, hipEventDisableTiming));
        entry.events.push_back(er);
        TMA_HIP_CHECK(hipEventRecord(er, streams_[r].stream()));
        TMA_HIP_CHECK(hipStreamWaitEvent(s0.stream(), er, 0));
      }
      hipGraph_t graph = nullptr;
      TMA_HIP_CHECK(hipStreamEndCapture(s0.stream(), &graph));
      capturing = false;
      TMA_HIP_CHECK(hipGraphInstantiate(&entry.exec, graph, nullptr, nullptr, 0));
      TMA_HIP_CHECK(hipGraphDestroy(graph));
      graphs_[GraphKey{base, count, static_cast<int>(d), static_cast<int>(op),
                       alpha, nrings}] = entry;
    } catch (const std::exception& e) {
      if (capturing) {
        hipGraph_t g = nullptr;
        (void)hipStreamEndCapture(s0.stream(), &g);
        if (g) (void)hipGraphDestroy(g);
      }
      for (auto ev : entry.events) (void)hipEventDestroy(ev);
      graphs_enabled_ = false;  // never retry; eager path is always correct
      if (std::getenv("TORCHMPI_LOG")) {
        fprintf(stderr, "[torchmpi_amd] hipGraph capture disabled: %s\n",
                e.what());
      }
    }
  }

  int rank_, world_, device_;
  std::vector<std::vector<int>> rings_;
  std::vector<ncclComm_t> comms_;
  std::vector<c10::hip::HIPStreamMasqueradingAsCUDA> streams_;
  std::vector<int> pos_, next_, prev_;
  int used_rings() const { return std::min(num_rings(), max_rings_); }

  int max_rings_ = 1 << 30;
  bool ring_overlap_ = false;
  std::vector<hipEvent_t> fork_ev_;
  std::vector<std::vector<hipEvent_t>> recv_done_, red_done_;
  std::vector<c10::hip::HIPStreamMasqueradingAsCUDA> recv_streams_, red_streams_;
  hipEvent_t order_ev_ = nullptr;
  hipEvent_t join_ev_ = nullptr;
  size_t chunk_bytes_;
  size_t multiring_min_bytes_;
  size_t rhd_max_bytes_ = 256u << 10;
  bool graphs_enabled_ = true;
  size_t graph_min_bytes_ = 1u << 20;
  std::unordered_map<GraphKey, GraphEntry, GraphKeyHash> graphs_;
  std::map<size_t, at::Tensor> graph_scratch_pool_;
};

// ---------------------------------------------------------------------------
// Kernel entry points exposed to Python (for numerics tests and the bucket
// engine).
// ---------------------------------------------------------------------------

namespace {

void py_reduce_into(at::Tensor dst, at::Tensor src, int op) {
  check_gpu_tensor(dst);
  check_gpu_tensor(src);
  TMA_CHECK(dst.numel() == src.numel() && dst.scalar_type() == src.scalar_type(),
            "reduce_into: mismatched tensors");
  launch_reduce(dst.data_ptr(), src.data_ptr(), dst.numel(), dtype_of(dst),
                static_cast<RedOp>(op), c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream());
}

void py_scale(at::Tensor t, double alpha) {
  check_gpu_tensor(t);
  launch_scale(t.data_ptr(), t.numel(), dtype_of(t), alpha,
               c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream());
}

struct HostPackMeta {
  void* ptr;
  unsigned long long off;
  unsigned long long n;
};

// Build the device-side meta table for a bucket: tensors[i] lives at
// offsets[i] (in bucket elements). Returns a persistent device byte tensor.
at::Tensor py_build_pack_meta(std::vector<at::Tensor> tensors,
                              std::vector<int64_t> offsets) {
  TMA_CHECK(tensors.size() == offsets.size(), "meta: size mismatch");
  TMA_CHECK(!tensors.empty(), "meta: empty tensor list");
  auto host = at::empty({static_cast<int64_t>(tensors.size() * sizeof(HostPackMeta))},
                        at::TensorOptions().dtype(at::kByte));
  auto* metas = reinterpret_cast<HostPackMeta*>(host.data_ptr());
  for (size_t i = 0; i < tensors.size(); ++i) {
    // Dense non-overlapping is enough (channels_last grads etc.): pack copies
    // in raw storage order, identical on every rank.
    TMA_CHECK(tensors[i].is_cuda(), "pack: expected GPU tensors");
    TMA_CHECK(tensors[i].is_non_overlapping_and_dense(),
              "pack: tensors must be dense");
    metas[i].ptr = tensors[i].data_ptr();
    metas[i].off = static_cast<unsigned long long>(offsets[i]);
    metas[i].n = static_cast<unsigned long long>(tensors[i].numel());
  }
  return host.to(tensors[0].device(), /*non_blocking=*/false);
}

void py_pack(at::Tensor bucket, at::Tensor meta_dev, int64_t ntensors,
             int64_t tensor_dtype) {
  check_gpu_tensor(bucket);
  launch_pack(bucket.data_ptr(), dtype_of(bucket), meta_dev.data_ptr(),
              static_cast<int>(ntensors), bucket.numel(),
              static_cast<DType>(tensor_dtype),
              c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream());
}

void py_unpack(at::Tensor bucket, at::Tensor meta_dev, int64_t ntensors,
               int64_t tensor_dtype, double alpha) {
  check_gpu_tensor(bucket);
  launch_unpack(bucket.data_ptr(), dtype_of(bucket), meta_dev.data_ptr(),
                static_cast<int>(ntensors), bucket.numel(),
                static_cast<DType>(tensor_dtype), alpha,
                c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream());
}

std::string py_get_unique_id() {
  ncclUniqueId id;
  TMA_NCCL_CHECK(ncclGetUniqueId(&id));
  return std::string(reinterpret_cast<char*>(&id), sizeof(id));
}

// The exact ring schedules the engine executes, exposed for simulation tests
// (the 8-GPU path cannot run on the 1-GPU CI box; tests/test_ring_schedule.py
// simulates these step lists and checks the allreduce/reduce-scatter
// invariants for every world size x ring permutation).
// phase 0 = reduce-scatter, phase 1 = allgather.
// Returns steps x world x 2: [step][position] = (send_seg, recv_seg).
std::vector<std::vector<std::vector<int>>> py_ring_schedule(
    int world, std::vector<int> perm, int phase) {
  const int W = world;
  std::vector<std::vector<std::vector<int>>> out;
  for (int step = 0; step < W - 1; ++step) {
    std::vector<std::vector<int>> row;
    for (int p = 0; p < W; ++p) {
      int send_seg, recv_seg;
      ring_step_segs(perm, W, p, step, phase, send_seg, recv_seg);
      row.push_back({send_seg, recv_seg});
    }
    out.push_back(row);
  }
  return out;
}

}  // namespace

void register_gpu(pybind11::module_& m) {
  namespace py = pybind11;

  py::class_<GpuWork, std::shared_ptr<GpuWork>>(m, "GpuWork")
      .def("wait", &GpuWork::wait)
      .def("wait_host", &GpuWork::wait_host,
           py::call_guard<py::gil_scoped_release>())
      .def("query", &GpuWork::query);

  py::class_<RcclEngine>(m, "RcclEngine")
      .def(py::init([](int rank, int world, int device, std::vector<py::bytes> uids,
                       std::vector<std::vector<int>> rings) {
             std::vector<std::string> u;
             for (auto& b : uids) u.push_back(std::string(b));
             pybind11::gil_scoped_release nogil;
             return std::make_unique<RcclEngine>(rank, world, device, u, rings);
           }),
           py::arg("rank"), py::arg("world"), py::arg("device"), py::arg("uids"),
           py::arg("rings"))
      .def("rank", &RcclEngine::rank)
      .def("size", &RcclEngine::size)
      .def("num_rings", &RcclEngine::num_rings)
      .def("graphs_enabled", &RcclEngine::graphs_enabled)
      .def("set_graphs_enabled", &RcclEngine::set_graphs_enabled)
      .def("set_max_rings", &RcclEngine::set_max_rings)
      .def("graph_selfcheck", &RcclEngine::graph_selfcheck,
           py::call_guard<py::gil_scoped_release>())
      .def("allreduce", &RcclEngine::allreduce, py::arg("tensor"), py::arg("op"),
           py::arg("scale_alpha") = 1.0, py::call_guard<py::gil_scoped_release>())
      .def("reduce_scatter", &RcclEngine::reduce_scatter,
           py::call_guard<py::gil_scoped_release>())
      .def("broadcast", &RcclEngine::broadcast,
           py::call_guard<py::gil_scoped_release>())
      .def("reduce", &RcclEngine::reduce, py::call_guard<py::gil_scoped_release>())
      .def("allgather", &RcclEngine::allgather,
           py::call_guard<py::gil_scoped_release>())
      .def("sendrecv", &RcclEngine::sendrecv,
           py::call_guard<py::gil_scoped_release>())
      .def("gather", &RcclEngine::gather,
           py::call_guard<py::gil_scoped_release>())
      .def("scatter", &RcclEngine::scatter,
           py::call_guard<py::gil_scoped_release>())
      .def("alltoall", &RcclEngine::alltoall,
           py::call_guard<py::gil_scoped_release>())
      .def("send", &RcclEngine::send, py::call_guard<py::gil_scoped_release>())
      .def("recv", &RcclEngine::recv, py::call_guard<py::gil_scoped_release>());

  m.def("nccl_unique_id", [] { return pybind11::bytes(py_get_unique_id()); });
  m.def("ring_schedule", &py_ring_schedule,
        "engine ring schedule for simulation tests");
  m.def("segment_parts",
        [](std::vector<size_t> off, std::vector<size_t> cnt, int r, int nrings,
           size_t align) {
          // mirrors RcclEngine::segment_parts (same split_aligned core)
          std::vector<size_t> off_r(off.size()), cnt_r(off.size());
          std::vector<size_t> po, pc;
          for (size_t g = 0; g < off.size(); ++g) {
            split_aligned(cnt[g], nrings, align, po, pc);
            off_r[g] = off[g] + po[r];
            cnt_r[g] = pc[r];
          }
          return std::make_pair(off_r, cnt_r);
        },
        "per-ring segment part tables (multi-ring verbs), for tests");
  m.def("reduce_into", &py_reduce_into, "dst = dst (op) src on current stream");
  m.def("scale_", &py_scale);
  m.def("build_pack_meta", &py_build_pack_meta);
  m.def("pack_bucket", &py_pack);
  m.def("unpack_bucket", &py_unpack);
  m.def("hip_device_count", [] {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    return e == hipSuccess ? n : 0;
  });
}

}  // namespace tma
