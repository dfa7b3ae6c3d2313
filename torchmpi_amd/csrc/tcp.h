// TCP socket mesh: the CPU transport's wire layer.
//
// MI355X-native replacement for the reference's MPI point-to-point transport
// (TorchMPI lib/collectives.cpp MPI_Isend/Irecv paths):
// a full mesh of non-blocking TCP connections between ranks, with poll()-based
// duplex transfers so ring steps (recv-from-left while send-to-right) cannot
// deadlock.
#pragma once

#include <cstddef>
#include <cstdint>
#include <string>
#include <vector>

namespace tma {

// One endpoint of the mesh. Sockets are kept non-blocking.
class TcpMesh {
 public:
  // Create a listening socket on bind_addr (host only; ephemeral port).
  // Call listen_port() to learn the port, exchange addresses out-of-band
  // (Python TCPStore), then connect_mesh().
  explicit TcpMesh(const std::string& bind_addr);
  ~TcpMesh();

  TcpMesh(const TcpMesh&) = delete;
  TcpMesh& operator=(const TcpMesh&) = delete;

  int listen_port() const { return port_; }

  // addrs[i] = "host:port" for rank i. Ranks > our rank connect to us; we
  // connect to ranks < our rank. Populates the fd table.
  void connect_mesh(int rank, const std::vector<std::string>& addrs);

  // Blocking (via poll) complete send / recv of exactly n bytes.
  void send_all(int peer, const void* buf, size_t n);
  void recv_all(int peer, void* buf, size_t n);

  // Simultaneously send sn bytes to speer and receive rn bytes from rpeer.
  // speer/rpeer may be the same rank (full-duplex on one socket) and either
  // may be -1 (skip that direction).
  void duplex(int speer, const void* sbuf, size_t sn, int rpeer, void* rbuf, size_t rn);

  int world_size() const { return static_cast<int>(fds_.size()); }
  int rank() const { return rank_; }
  // Raw fd for peer (for external poll loops, e.g. the parameter server).
  int fd(int peer) const { return fds_[peer]; }

 private:
  int listen_fd_ = -1;
  int port_ = 0;
  int rank_ = -1;
  std::vector<int> fds_;  // fds_[peer]; own entry = -1
};

}  // namespace tma
