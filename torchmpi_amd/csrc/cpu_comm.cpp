#include "cpu_comm.h"

#include <cstdlib>
#include <cstring>
#include <vector>

namespace tma {

void Work::wait() {
  std::unique_lock<std::mutex> lk(m);
  cv.wait(lk, [&] { return done; });
  if (err) std::rethrow_exception(err);
}

bool Work::ready() {
  std::lock_guard<std::mutex> lk(m);
  return done;
}

void Work::finish(std::exception_ptr e) {
  {
    std::lock_guard<std::mutex> lk(m);
    done = true;
    err = e;
  }
  cv.notify_all();
}

CpuComm::CpuComm(int rank, const std::string& bind_addr) : rank_(rank), mesh_(bind_addr) {
  const char* cb = std::getenv("TORCHMPI_CHUNK_BYTES");
  chunk_bytes_ = cb ? static_cast<size_t>(std::atoll(cb)) : (1u << 20);
  if (chunk_bytes_ < 4096) chunk_bytes_ = 4096;
  const char* rm = std::getenv("TORCHMPI_RHD_MAX_BYTES");
  rhd_max_bytes_ = rm ? static_cast<size_t>(std::atoll(rm)) : (256u << 10);
  worker_ = std::thread([this] { worker_loop(); });
}

void CpuComm::connect(int rank, const std::vector<std::string>& addrs) {
  rank_ = rank;
  world_ = static_cast<int>(addrs.size());
  if (world_ > 1) mesh_.connect_mesh(rank, addrs);
}

CpuComm::~CpuComm() {
  {
    std::lock_guard<std::mutex> lk(qm_);
    stop_ = true;
  }
  qcv_.notify_all();
  if (worker_.joinable()) worker_.join();
}

std::shared_ptr<Work> CpuComm::enqueue(std::function<void()> fn) {
  auto w = std::make_shared<Work>();
  {
    std::lock_guard<std::mutex> lk(qm_);
    TMA_CHECK(!stop_, "communicator already stopped");
    queue_.push_back({std::move(fn), w});
  }
  qcv_.notify_one();
  return w;
}

void CpuComm::worker_loop() {
  for (;;) {
    Task t;
    {
      std::unique_lock<std::mutex> lk(qm_);
      qcv_.wait(lk, [&] { return stop_ || !queue_.empty(); });
      if (queue_.empty()) return;  // stop_ and drained
      t = std::move(queue_.front());
      queue_.pop_front();
    }
    std::exception_ptr err;
    try {
      t.fn();
    } catch (...) {
      err = std::current_exception();
    }
    t.work->finish(err);
  }
}

// ---------------------------------------------------------------------------
// Collectives
// ---------------------------------------------------------------------------

namespace {
inline bool is_pow2(int x) { return x > 0 && (x & (x - 1)) == 0; }
}  // namespace

// Recursive halving-doubling allreduce: log2(W) exchange steps (vs the
// ring's 2(W-1)) — the latency algorithm for small payloads. Requires W a
// power of two. Segment ownership after the halving phase is bit-reversed,
// mirrored back by the doubling phase.
void CpuComm::allreduce_rhd(void* data, size_t count, DType d, RedOp op) {
  const size_t es = dtype_size(d);
  char* base = static_cast<char*>(data);
  const int W = world_;
  std::vector<size_t> off, cnt;
  split_segments(count, W, off, cnt);
  auto range_bytes = [&](int s0, int s1, size_t& o, size_t& n) {
    o = off[s0] * es;
    n = (off[s1 - 1] + cnt[s1 - 1] - off[s0]) * es;
  };
  std::vector<char> scratch(count * es);
  int logw = 0;
  while ((1 << logw) < W) ++logw;
  int s0 = 0, s1 = W;
  // Halving phase: keep one half, exchange + reduce.
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
    mesh_.duplex(partner, base + so, sn, partner, scratch.data(), kn);
    if (kn > 0)
      cpu_reduce(base + ko, scratch.data(), kn / es, d, op);
    s0 = keep0; s1 = keep1;
  }
  // Doubling phase: mirror the exchanges back.
  for (int k = logw - 1; k >= 0; --k) {
    int partner = rank_ ^ (1 << k);
    // partner's owned range at this level is the sibling of ours.
    int width = s1 - s0;
    int sib0 = (s0 % (2 * width) == 0) ? s0 + width : s0 - width;
    int sib1 = sib0 + width;
    size_t mo, mn, po, pn;
    range_bytes(s0, s1, mo, mn);
    range_bytes(sib0, sib1, po, pn);
    mesh_.duplex(partner, base + mo, mn, partner, base + po, pn);
    s0 = std::min(s0, sib0);
    s1 = std::max(s1, sib1);
  }
}

void CpuComm::allreduce(void* data, size_t count, DType d, RedOp op) {
  if (world_ == 1 || count == 0) return;
  const size_t es = dtype_size(d);
  if (is_pow2(world_) && count * es <= rhd_max_bytes_ &&
      count >= static_cast<size_t>(world_)) {
    allreduce_rhd(data, count, d, op);
    return;
  }
  char* base = static_cast<char*>(data);
  const int W = world_;
  const int right = (rank_ + 1) % W, left = (rank_ - 1 + W) % W;

  std::vector<size_t> off, cnt;
  split_segments(count, W, off, cnt);
  size_t max_seg = cnt[0];
  std::vector<char> scratch(max_seg * es);

  // Phase 1: ring reduce-scatter. Schedule sends segment (r - s - 1) so that
  // after W-1 steps rank r owns the fully reduced segment r.
  for (int s = 0; s < W - 1; ++s) {
    int send_seg = (rank_ - s - 1 + 2 * W) % W;
    int recv_seg = (rank_ - s - 2 + 2 * W) % W;
    mesh_.duplex(right, base + off[send_seg] * es, cnt[send_seg] * es, left,
                 scratch.data(), cnt[recv_seg] * es);
    cpu_reduce(base + off[recv_seg] * es, scratch.data(), cnt[recv_seg], d, op);
  }
  // Phase 2: ring allgather of the reduced segments (rank r starts owning r).
  for (int s = 0; s < W - 1; ++s) {
    int send_seg = (rank_ - s + W) % W;
    int recv_seg = (rank_ - s - 1 + 2 * W) % W;
    mesh_.duplex(right, base + off[send_seg] * es, cnt[send_seg] * es, left,
                 base + off[recv_seg] * es, cnt[recv_seg] * es);
  }
}

void CpuComm::reduce_scatter(void* data, void* out, size_t total_count, DType d,
                             RedOp op) {
  const size_t es = dtype_size(d);
  std::vector<size_t> off, cnt;
  split_segments(total_count, world_, off, cnt);
  char* base = static_cast<char*>(data);
  if (world_ == 1) {
    if (out != data) std::memcpy(out, data, total_count * es);
    return;
  }
  const int W = world_;
  const int right = (rank_ + 1) % W, left = (rank_ - 1 + W) % W;
  std::vector<char> scratch(cnt[0] * es);
  for (int s = 0; s < W - 1; ++s) {
    int send_seg = (rank_ - s - 1 + 2 * W) % W;
    int recv_seg = (rank_ - s - 2 + 2 * W) % W;
    mesh_.duplex(right, base + off[send_seg] * es, cnt[send_seg] * es, left,
                 scratch.data(), cnt[recv_seg] * es);
    cpu_reduce(base + off[recv_seg] * es, scratch.data(), cnt[recv_seg], d, op);
  }
  std::memcpy(out, base + off[rank_] * es, cnt[rank_] * es);
}

void CpuComm::broadcast(void* data, size_t bytes, int root) {
  if (world_ == 1 || bytes == 0) return;
  // Pipelined chain ordered from the root.
  const int W = world_;
  int pos = (rank_ - root + W) % W;
  int prev = pos > 0 ? (rank_ - 1 + W) % W : -1;
  int next = pos < W - 1 ? (rank_ + 1) % W : -1;
  char* base = static_cast<char*>(data);

  size_t nchunks = (bytes + chunk_bytes_ - 1) / chunk_bytes_;
  auto chunk_at = [&](size_t c, size_t& o, size_t& n) {
    o = c * chunk_bytes_;
    n = o + chunk_bytes_ <= bytes ? chunk_bytes_ : bytes - o;
  };
  // Middle ranks receive chunk c while forwarding chunk c-1.
  for (size_t c = 0; c <= nchunks; ++c) {
    size_t ro = 0, rn = 0, so = 0, sn = 0;
    bool do_recv = prev >= 0 && c < nchunks;
    bool do_send = next >= 0 && c > 0;
    if (do_recv) chunk_at(c, ro, rn);
    if (do_send) chunk_at(c - 1, so, sn);
    if (do_recv || do_send) {
      mesh_.duplex(do_send ? next : -1, base + so, do_send ? sn : 0,
                   do_recv ? prev : -1, base + ro, do_recv ? rn : 0);
    }
  }
}

void CpuComm::reduce(void* data, size_t count, DType d, RedOp op, int root) {
  if (world_ == 1 || count == 0) return;
  const size_t es = dtype_size(d);
  const int W = world_;
  int rel = (rank_ - root + W) % W;
  // Binomial tree: accumulate into a private copy so non-root inputs are
  // preserved (MPI_Reduce semantics; root's buffer receives the result).
  std::vector<char> acc(count * es);
  std::memcpy(acc.data(), data, count * es);
  std::vector<char> scratch(count * es);
  for (int mask = 1; mask < W; mask <<= 1) {
    if (rel & mask) {
      int parent = ((rel & ~mask) + root) % W;
      mesh_.send_all(parent, acc.data(), count * es);
      return;
    }
    int child_rel = rel | mask;
    if (child_rel < W) {
      int child = (child_rel + root) % W;
      mesh_.recv_all(child, scratch.data(), count * es);
      cpu_reduce(acc.data(), scratch.data(), count, d, op);
    }
  }
  std::memcpy(data, acc.data(), count * es);
}

void CpuComm::allgather(const void* in, void* out, size_t bytes_per_rank) {
  char* ob = static_cast<char*>(out);
  std::memcpy(ob + rank_ * bytes_per_rank, in, bytes_per_rank);
  if (world_ == 1) return;
  const int W = world_;
  const int right = (rank_ + 1) % W, left = (rank_ - 1 + W) % W;
  for (int s = 0; s < W - 1; ++s) {
    int send_slot = (rank_ - s + W) % W;
    int recv_slot = (rank_ - s - 1 + W) % W;
    mesh_.duplex(right, ob + send_slot * bytes_per_rank, bytes_per_rank, left,
                 ob + recv_slot * bytes_per_rank, bytes_per_rank);
  }
}

void CpuComm::gather(const void* in, void* out, size_t bytes_per_rank, int root) {
  if (rank_ == root) {
    char* ob = static_cast<char*>(out);
    std::memcpy(ob + rank_ * bytes_per_rank, in, bytes_per_rank);
    for (int r = 0; r < world_; ++r)
      if (r != root) mesh_.recv_all(r, ob + r * bytes_per_rank, bytes_per_rank);
  } else {
    mesh_.send_all(root, in, bytes_per_rank);
  }
}

void CpuComm::scatter(const void* in, void* out, size_t bytes_per_rank, int root) {
  if (rank_ == root) {
    const char* ib = static_cast<const char*>(in);
    for (int r = 0; r < world_; ++r) {
      if (r == root)
        std::memcpy(out, ib + r * bytes_per_rank, bytes_per_rank);
      else
        mesh_.send_all(r, ib + r * bytes_per_rank, bytes_per_rank);
    }
  } else {
    mesh_.recv_all(root, out, bytes_per_rank);
  }
}

void CpuComm::alltoall(const void* in, void* out, size_t bytes_per_pair) {
  const char* ib = static_cast<const char*>(in);
  char* ob = static_cast<char*>(out);
  std::memcpy(ob + rank_ * bytes_per_pair, ib + rank_ * bytes_per_pair, bytes_per_pair);
  // Pairwise exchange rounds: partner = rank XOR-free rotation.
  for (int s = 1; s < world_; ++s) {
    int dst = (rank_ + s) % world_;
    int src = (rank_ - s + world_) % world_;
    mesh_.duplex(dst, ib + dst * bytes_per_pair, bytes_per_pair, src,
                 ob + src * bytes_per_pair, bytes_per_pair);
  }
}

void CpuComm::sendrecv(const void* sbuf, size_t sbytes, int dst, void* rbuf,
                       size_t rbytes, int src) {
  if (dst == rank_ && src == rank_) {
    std::memcpy(rbuf, sbuf, rbytes < sbytes ? rbytes : sbytes);
    return;
  }
  mesh_.duplex(dst, sbuf, sbytes, src, rbuf, rbytes);
}

void CpuComm::send(const void* buf, size_t bytes, int dst) {
  mesh_.send_all(dst, buf, bytes);
}

void CpuComm::recv(void* buf, size_t bytes, int src) { mesh_.recv_all(src, buf, bytes); }

void CpuComm::barrier() {
  if (world_ == 1) return;
  // Dissemination barrier: log2(W) rounds.
  char tok = 1, got = 0;
  for (int k = 1; k < world_; k <<= 1) {
    int dst = (rank_ + k) % world_;
    int src = (rank_ - k + world_) % world_;
    mesh_.duplex(dst, &tok, 1, src, &got, 1);
  }
}

}  // namespace tma
