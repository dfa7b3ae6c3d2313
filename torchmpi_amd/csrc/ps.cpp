#include "ps.h"

#include <errno.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>

namespace tma {

namespace {

enum FrameType : uint8_t { kUpdate = 1, kFetch = 2, kAck = 3, kData = 4 };

struct FrameHdr {
  uint8_t type;
  uint8_t rule;   // 0 = copy, 1 = add
  uint8_t dtype;
  uint8_t pad;
  int32_t tid;
  uint64_t seq;
  uint64_t bytes;  // payload bytes following the header
};
static_assert(sizeof(FrameHdr) == 24, "frame header size");

}  // namespace

PsContext::PsContext(int rank, const std::string& bind_addr)
    : rank_(rank), mesh_(bind_addr) {}

PsContext::~PsContext() { stop(); }

void PsContext::start(int rank, const std::vector<std::string>& addrs) {
  rank_ = rank;
  world_ = static_cast<int>(addrs.size());
  if (world_ > 1) mesh_.connect_mesh(rank, addrs);
  outq_.clear();
  outq_.resize(world_);
  in_.clear();
  in_.resize(world_);
  for (auto& st : in_) st.buf.resize(sizeof(FrameHdr));
  out_bytes_ = 0;
  dead_ = false;
  TMA_CHECK(pipe(wake_pipe_) == 0, "pipe failed");
  running_ = true;
  if (world_ > 1) loop_thread_ = std::thread([this] { loop(); });
}

void PsContext::stop() {
  if (!running_.exchange(false)) return;
  if (wake_pipe_[1] >= 0) {
    char b = 1;
    (void)!write(wake_pipe_[1], &b, 1);
  }
  if (loop_thread_.joinable()) loop_thread_.join();
  for (int i = 0; i < 2; ++i)
    if (wake_pipe_[i] >= 0) {
      close(wake_pipe_[i]);
      wake_pipe_[i] = -1;
    }
  // Anything still pending can never complete once the loop is gone.
  fail_all("parameter server stopped");
}

void PsContext::fail_all(const std::string& why) {
  dead_ = true;
  std::unordered_map<uint64_t, std::shared_ptr<Pending>> doomed;
  {
    std::lock_guard<std::mutex> lk(pend_m_);
    doomed.swap(pending_);
  }
  for (auto& kv : doomed) {
    kv.second->work->finish(std::make_exception_ptr(
        std::runtime_error("torchmpi_amd ps: " + why)));
  }
  out_cv_.notify_all();  // release backpressure waiters (they re-check dead_)
}

void PsContext::enqueue_frame(int peer, const void* hdr, size_t hdr_n,
                              const void* payload, size_t payload_n,
                              bool backpressure) {
  std::vector<char> frame(hdr_n + payload_n);
  std::memcpy(frame.data(), hdr, hdr_n);
  if (payload_n > 0)
    std::memcpy(frame.data() + hdr_n, payload, payload_n);
  {
    std::unique_lock<std::mutex> lk(out_m_);
    if (backpressure) {
      out_cv_.wait(lk, [&] {
        return dead_.load() || !running_.load() || out_bytes_ < kMaxQueuedBytes;
      });
      TMA_CHECK(!dead_.load() && running_.load(),
                "parameter server event loop is not running");
    }
    out_bytes_ += frame.size();
    outq_[peer].frames.push_back(std::move(frame));
  }
  if (wake_pipe_[1] >= 0) {
    char b = 1;
    (void)!write(wake_pipe_[1], &b, 1);
  }
}

bool PsContext::drain_out(int peer) {
  int fd = mesh_fd(peer);
  while (true) {
    std::vector<char>* front;
    size_t off;
    {
      std::lock_guard<std::mutex> lk(out_m_);
      auto& q = outq_[peer];
      if (q.frames.empty()) return true;
      front = &q.frames.front();
      off = q.head_off;
    }
    // Only this thread pops/writes, so the front pointer stays valid
    // outside the lock; enqueuers only push_back.
    ssize_t n = ::send(fd, front->data() + off, front->size() - off, MSG_NOSIGNAL);
    if (n < 0) {
      if (errno == EINTR) continue;
      if (errno == EAGAIN || errno == EWOULDBLOCK) return true;  // socket full
      return false;
    }
    std::lock_guard<std::mutex> lk(out_m_);
    auto& q = outq_[peer];
    q.head_off += static_cast<size_t>(n);
    out_bytes_ -= static_cast<size_t>(n);
    if (q.head_off == q.frames.front().size()) {
      q.frames.pop_front();
      q.head_off = 0;
    }
    out_cv_.notify_all();
  }
}

PsContext::Reg& PsContext::reg(int tid) {
  std::lock_guard<std::mutex> lk(regs_m_);
  auto it = regs_.find(tid);
  TMA_CHECK(it != regs_.end(), "parameterserver: unregistered tensor id");
  return *it->second;
}

void PsContext::register_tensor(int tid, size_t count, DType d, const void* init) {
  auto r = std::make_unique<Reg>();
  r->count = count;
  r->dtype = d;
  split_segments(count, world_, r->off, r->cnt);
  size_t es = dtype_size(d);
  r->shard.resize(r->cnt[rank_] * es);
  std::memcpy(r->shard.data(),
              static_cast<const char*>(init) + r->off[rank_] * es,
              r->cnt[rank_] * es);
  std::lock_guard<std::mutex> lk(regs_m_);
  regs_[tid] = std::move(r);
}

void PsContext::free_tensor(int tid) {
  std::lock_guard<std::mutex> lk(regs_m_);
  regs_.erase(tid);
}

void PsContext::shard_of(int tid, int r, size_t& off, size_t& cnt) {
  Reg& g = reg(tid);
  off = g.off[r];
  cnt = g.cnt[r];
}

std::shared_ptr<Work> PsContext::send(int tid, const void* data, int rule) {
  Reg& g = reg(tid);
  size_t es = dtype_size(g.dtype);
  auto pend = std::make_shared<Pending>();
  pend->work = std::make_shared<Work>();
  uint64_t seq = next_seq_.fetch_add(1);
  // Count remote owners first so an early ACK cannot complete prematurely.
  int remotes = 0;
  for (int r = 0; r < world_; ++r)
    if (r != rank_ && g.cnt[r] > 0) ++remotes;
  pend->remaining = remotes;
  if (remotes > 0) {
    std::lock_guard<std::mutex> lk(pend_m_);
    pending_[seq] = pend;
  }
  // Local shard: apply directly.
  if (g.cnt[rank_] > 0) {
    std::lock_guard<std::mutex> lk(g.m);
    const char* src = static_cast<const char*>(data) + g.off[rank_] * es;
    if (rule == 0)
      std::memcpy(g.shard.data(), src, g.cnt[rank_] * es);
    else
      cpu_reduce(g.shard.data(), src, g.cnt[rank_], g.dtype, RedOp::SUM);
  }
  // Remote shards.
  for (int r = 0; r < world_; ++r) {
    if (r == rank_ || g.cnt[r] == 0) continue;
    FrameHdr h{kUpdate, static_cast<uint8_t>(rule),
               static_cast<uint8_t>(static_cast<int>(g.dtype)), 0, tid, seq,
               g.cnt[r] * es};
    enqueue_frame(r, &h, sizeof(h),
                  static_cast<const char*>(data) + g.off[r] * es,
                  g.cnt[r] * es, /*backpressure=*/true);
  }
  if (remotes == 0) pend->work->finish(nullptr);
  return pend->work;
}

std::shared_ptr<Work> PsContext::receive(int tid, void* data) {
  Reg& g = reg(tid);
  size_t es = dtype_size(g.dtype);
  auto pend = std::make_shared<Pending>();
  pend->work = std::make_shared<Work>();
  pend->dest = data;
  pend->tid = tid;
  uint64_t seq = next_seq_.fetch_add(1);
  int remotes = 0;
  for (int r = 0; r < world_; ++r)
    if (r != rank_ && g.cnt[r] > 0) ++remotes;
  pend->remaining = remotes;
  if (remotes > 0) {
    std::lock_guard<std::mutex> lk(pend_m_);
    pending_[seq] = pend;
  }
  if (g.cnt[rank_] > 0) {
    std::lock_guard<std::mutex> lk(g.m);
    std::memcpy(static_cast<char*>(data) + g.off[rank_] * es, g.shard.data(),
                g.cnt[rank_] * es);
  }
  for (int r = 0; r < world_; ++r) {
    if (r == rank_ || g.cnt[r] == 0) continue;
    FrameHdr h{kFetch, 0, 0, 0, tid, seq, 0};
    enqueue_frame(r, &h, sizeof(h), nullptr, 0, /*backpressure=*/true);
  }
  if (remotes == 0) pend->work->finish(nullptr);
  return pend->work;
}

bool PsContext::read_in(int peer) {
  int fd = mesh_fd(peer);
  auto& st = in_[peer];
  while (true) {
    ssize_t n = ::recv(fd, st.buf.data() + st.filled,
                       st.buf.size() - st.filled, MSG_DONTWAIT);
    if (n == 0) return false;  // peer closed
    if (n < 0) {
      if (errno == EINTR) continue;
      if (errno == EAGAIN || errno == EWOULDBLOCK) return true;
      return false;
    }
    st.filled += static_cast<size_t>(n);
    if (st.filled < st.buf.size()) continue;
    if (!st.have_hdr) {
      FrameHdr h;
      std::memcpy(&h, st.buf.data(), sizeof(h));
      st.have_hdr = true;
      if (h.bytes > 0) {
        st.buf.resize(sizeof(FrameHdr) + h.bytes);
        std::memcpy(st.buf.data(), &h, sizeof(h));
        continue;  // keep filling until the payload is complete
      }
    }
    dispatch(peer, st.buf.data(), st.buf.data() + sizeof(FrameHdr));
    st.buf.assign(sizeof(FrameHdr), 0);
    st.filled = 0;
    st.have_hdr = false;
  }
}

void PsContext::dispatch(int peer, const void* hdr, const char* payload) {
  FrameHdr h;
  std::memcpy(&h, hdr, sizeof(h));
  switch (h.type) {
    case kUpdate: {
      Reg& g = reg(h.tid);
      {
        std::lock_guard<std::mutex> lk(g.m);
        TMA_CHECK(h.bytes == g.shard.size(), "update size mismatch");
        if (h.rule == 0)
          std::memcpy(g.shard.data(), payload, h.bytes);
        else
          cpu_reduce(g.shard.data(), payload, g.cnt[rank_], g.dtype, RedOp::SUM);
      }
      FrameHdr ack{kAck, 0, 0, 0, h.tid, h.seq, 0};
      enqueue_frame(peer, &ack, sizeof(ack), nullptr, 0, /*backpressure=*/false);
      break;
    }
    case kFetch: {
      Reg& g = reg(h.tid);
      std::vector<char> snapshot;
      {
        std::lock_guard<std::mutex> lk(g.m);
        snapshot = g.shard;
      }
      FrameHdr d{kData, 0, static_cast<uint8_t>(static_cast<int>(g.dtype)), 0, h.tid,
                 h.seq, snapshot.size()};
      enqueue_frame(peer, &d, sizeof(d), snapshot.data(), snapshot.size(),
                    /*backpressure=*/false);
      break;
    }
    case kAck: {
      std::shared_ptr<Pending> p;
      {
        std::lock_guard<std::mutex> lk(pend_m_);
        auto it = pending_.find(h.seq);
        if (it != pending_.end()) p = it->second;
      }
      if (p && p->remaining.fetch_sub(1) == 1) {
        {
          std::lock_guard<std::mutex> lk(pend_m_);
          pending_.erase(h.seq);
        }
        p->work->finish(nullptr);
      }
      break;
    }
    case kData: {
      std::shared_ptr<Pending> p;
      {
        std::lock_guard<std::mutex> lk(pend_m_);
        auto it = pending_.find(h.seq);
        if (it != pending_.end()) p = it->second;
      }
      TMA_CHECK(p != nullptr, "DATA reply with no pending receive");
      Reg& g = reg(h.tid);
      size_t es = dtype_size(g.dtype);
      TMA_CHECK(h.bytes == g.cnt[peer] * es, "DATA size mismatch");
      std::memcpy(static_cast<char*>(p->dest) + g.off[peer] * es, payload,
                  h.bytes);
      if (p->remaining.fetch_sub(1) == 1) {
        {
          std::lock_guard<std::mutex> lk(pend_m_);
          pending_.erase(h.seq);
        }
        p->work->finish(nullptr);
      }
      break;
    }
    default:
      TMA_CHECK(false, "bad PS frame type");
  }
}

void PsContext::loop() {
  std::vector<pollfd> fds;
  std::vector<int> peer_of;
  while (running_) {
    fds.clear();
    peer_of.clear();
    fds.push_back({wake_pipe_[0], POLLIN, 0});
    peer_of.push_back(-1);
    for (int r = 0; r < world_; ++r) {
      if (r == rank_) continue;
      // TcpMesh keeps one fd per peer; poll it for incoming frames, and
      // for writability whenever its out-queue holds bytes. This thread is
      // the mesh's only reader AND only writer (single-writer discipline),
      // so it can never block mid-frame: a stalled peer costs queue memory,
      // not liveness.
      short ev = POLLIN;
      {
        std::lock_guard<std::mutex> lk(out_m_);
        if (!outq_[r].frames.empty()) ev |= POLLOUT;
      }
      fds.push_back({mesh_fd(r), ev, 0});
      peer_of.push_back(r);
    }
    int rc = poll(fds.data(), fds.size(), 500);
    if (rc < 0) continue;  // EINTR: re-check running_
    if (fds[0].revents & POLLIN) {
      char b[64];
      (void)!read(wake_pipe_[0], b, sizeof(b));
    }
    for (size_t i = 1; i < fds.size(); ++i) {
      try {
        if (fds[i].revents & POLLOUT) {
          if (!drain_out(peer_of[i]))
            throw std::runtime_error("socket write error to peer " +
                                     std::to_string(peer_of[i]));
        }
        if (fds[i].revents & (POLLIN | POLLHUP | POLLERR)) {
          if (!read_in(peer_of[i]))
            throw std::runtime_error("socket read error/close from peer " +
                                     std::to_string(peer_of[i]));
        }
      } catch (const std::exception& e) {
        if (!running_) return;
        // Never std::terminate the process from the event loop: log, stop
        // serving, and fail every pending Work so wait() raises loudly.
        fprintf(stderr, "[torchmpi_amd ps] event loop error: %s\n", e.what());
        running_ = false;
        fail_all(std::string("event loop died: ") + e.what());
        return;
      }
    }
  }
}

}  // namespace tma
