// CPU communicator: TCP-mesh collectives + per-communicator async worker.
//
// MI355X-native re-design of TorchMPI's CPU collective layer
// (lib/collectives.cpp: sync MPI wrappers, chunked-pipelined allreducep2p /
// broadcastp2p, thread-pool async engine lib/thread_pool.h — SURVEY.md
// sections 2.1, 3.2). Algorithms here are ring reduce-scatter + allgather
// (allreduce), pipelined chain (broadcast), binomial tree (reduce),
// ring (allgather), dissemination (barrier) over raw TCP; async ops are
// executed in submission order by one worker thread per communicator, which
// is the reference's per-communicator ordering guarantee.
#pragma once

#include <condition_variable>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <thread>

#include "common.h"
#include "tcp.h"

namespace tma {

// Completion handle for an async operation.
struct Work {
  void wait();
  bool ready();
  void finish(std::exception_ptr e);

  std::mutex m;
  std::condition_variable cv;
  bool done = false;
  std::exception_ptr err;
};

class CpuComm {
 public:
  CpuComm(int rank, const std::string& bind_addr);
  ~CpuComm();

  int listen_port() const { return mesh_.listen_port(); }
  void connect(int rank, const std::vector<std::string>& addrs);

  int rank() const { return rank_; }
  int size() const { return world_; }

  // Enqueue fn for ordered execution on the worker thread.
  std::shared_ptr<Work> enqueue(std::function<void()> fn);

  // --- blocking verbs (call only from the worker thread via enqueue) ---
  void allreduce(void* data, size_t count, DType d, RedOp op);
  void allreduce_rhd(void* data, size_t count, DType d, RedOp op);
  void reduce_scatter(void* data, void* out, size_t total_count, DType d, RedOp op);
  void broadcast(void* data, size_t bytes, int root);
  void reduce(void* data, size_t count, DType d, RedOp op, int root);
  void allgather(const void* in, void* out, size_t bytes_per_rank);
  void gather(const void* in, void* out, size_t bytes_per_rank, int root);
  void scatter(const void* in, void* out, size_t bytes_per_rank, int root);
  void alltoall(const void* in, void* out, size_t bytes_per_pair);
  void sendrecv(const void* sbuf, size_t sbytes, int dst, void* rbuf, size_t rbytes,
                int src);
  void send(const void* buf, size_t bytes, int dst);
  void recv(void* buf, size_t bytes, int src);
  void barrier();

  TcpMesh& mesh() { return mesh_; }

 private:
  void worker_loop();

  int rank_ = 0;
  int world_ = 1;
  TcpMesh mesh_;
  size_t chunk_bytes_;
  size_t rhd_max_bytes_;

  struct Task {
    std::function<void()> fn;
    std::shared_ptr<Work> work;
  };
  std::thread worker_;
  std::deque<Task> queue_;
  std::mutex qm_;
  std::condition_variable qcv_;
  bool stop_ = false;
};

}  // namespace tma
