// Sharded parameter server over the TCP mesh.
//
// Re-design of TorchMPI's parameter server (lib/parameterserver.cpp,
// SURVEY.md sections 2.1 and 3.4): a registered tensor is sharded into
// world_size contiguous pieces, shard i owned by rank i. Clients push
// updates with a rule ('copy' = overwrite, 'add' = accumulate) and pull
// current values asynchronously; integer handles synchronize.
//
// Wire design (no MPI): the PS owns its own TCP mesh (separate from the
// collective mesh so PS traffic never interleaves with collectives). One
// event-loop thread per rank reads all peer sockets and dispatches frames:
// requests are served against the local shard store; replies complete
// pending client Works.
//
// Single-writer discipline: ALL outbound frames (client kUpdate/kFetch and
// loop-side kAck/kData) go through per-peer byte queues drained by the
// event loop with non-blocking writes under POLLOUT. The loop thread never
// blocks on a send, so two ranks simultaneously fetching each other's
// large shards cannot mutually stall (each loop keeps reading while its
// replies drain). Client enqueues are bounded (backpressure) and fail
// loudly if the loop has died, so Work.wait() raises instead of hanging.
// (This TCP PS serves CPU tensors; GPU tensors use the one-sided xGMI
// path in ipc_ps.hip and never touch this event loop.)
#pragma once

#include <atomic>
#include <condition_variable>
#include <deque>
#include <memory>
#include <mutex>
#include <thread>
#include <unordered_map>
#include <vector>

#include "common.h"
#include "cpu_comm.h"
#include "tcp.h"

namespace tma {

class PsContext {
 public:
  PsContext(int rank, const std::string& bind_addr);
  ~PsContext();

  int listen_port() const { return mesh_.listen_port(); }
  // Connect the PS mesh and start the event loop.
  void start(int rank, const std::vector<std::string>& addrs);
  void stop();

  // Register tensor `tid` of `count` elements; `init` points at the full
  // initial value (every rank passes the same count; each rank stores its
  // own shard from its slice of init).
  void register_tensor(int tid, size_t count, DType d, const void* init);
  void free_tensor(int tid);

  // Push `data` (full tensor) shard-wise to the owners with `rule`
  // (0 = copy, 1 = add). Completes when every owner has ACKed.
  std::shared_ptr<Work> send(int tid, const void* data, int rule);
  // Gather the current shard values into `data` (full tensor).
  std::shared_ptr<Work> receive(int tid, void* data);

  // Shard layout for tensor tid: offset/count (elements) of rank r's shard.
  void shard_of(int tid, int r, size_t& off, size_t& cnt);

 private:
  struct Reg {
    size_t count = 0;
    DType dtype = DType::F32;
    std::vector<char> shard;  // this rank's shard storage
    std::vector<size_t> off, cnt;
    std::mutex m;
  };

  struct Pending {
    std::shared_ptr<Work> work;
    std::atomic<int> remaining{0};
    void* dest = nullptr;  // for receive: full-tensor base pointer
    int tid = 0;
  };

  void loop();
  // Non-blocking read pump: accumulates bytes into the peer's frame buffer
  // and dispatches every complete frame; returns false on a fatal socket
  // error. The loop thread must never block on a read — a 16 MB kData
  // payload dribbles in across many poll iterations while THIS rank's own
  // replies keep draining (the round-1 advisor deadlock, both directions).
  bool read_in(int peer);
  void dispatch(int peer, const void* hdr, const char* payload);
  int mesh_fd(int peer) { return mesh_.fd(peer); }
  // Queue one frame (header + payload copied) for the loop thread to write.
  // backpressure=true (client threads) blocks while the queues hold more
  // than kMaxQueuedBytes; false (the loop's own replies) never blocks.
  void enqueue_frame(int peer, const void* hdr, size_t hdr_n, const void* payload,
                     size_t payload_n, bool backpressure);
  // Drain what the socket will take without blocking; returns false on a
  // fatal socket error.
  bool drain_out(int peer);
  // Mark the loop dead: fail every pending Work so wait() raises, wake
  // backpressure waiters, make further enqueues throw.
  void fail_all(const std::string& why);
  Reg& reg(int tid);

  int rank_ = 0;
  int world_ = 1;
  TcpMesh mesh_;
  std::thread loop_thread_;
  std::atomic<bool> running_{false};
  std::atomic<bool> dead_{false};
  int wake_pipe_[2] = {-1, -1};

  std::mutex regs_m_;
  std::unordered_map<int, std::unique_ptr<Reg>> regs_;

  std::mutex pend_m_;
  std::unordered_map<uint64_t, std::shared_ptr<Pending>> pending_;
  std::atomic<uint64_t> next_seq_{1};

  static constexpr size_t kMaxQueuedBytes = size_t(64) << 20;
  struct OutQ {
    std::deque<std::vector<char>> frames;
    size_t head_off = 0;  // bytes of frames.front() already written
  };
  std::mutex out_m_;
  std::condition_variable out_cv_;
  std::vector<OutQ> outq_;       // [peer]
  size_t out_bytes_ = 0;         // total queued across peers

  struct InSt {
    std::vector<char> buf;  // header, then header+payload
    size_t filled = 0;
    bool have_hdr = false;
  };
  std::vector<InSt> in_;         // [peer], loop thread only
};

}  // namespace tma
