#include "tcp.h"

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstdlib>
#include <cstring>
#include <stdexcept>

#include "common.h"

namespace tma {

namespace {

void set_nonblocking(int fd) {
  int flags = fcntl(fd, F_GETFL, 0);
  TMA_CHECK(flags >= 0, "fcntl F_GETFL failed");
  TMA_CHECK(fcntl(fd, F_SETFL, flags | O_NONBLOCK) == 0, "fcntl F_SETFL failed");
}

void set_sockopts(int fd) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  int buf = 4 << 20;
  setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &buf, sizeof(buf));
  setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &buf, sizeof(buf));
}

sockaddr_in parse_addr(const std::string& hostport) {
  auto colon = hostport.rfind(':');
  TMA_CHECK(colon != std::string::npos, "bad address " + hostport);
  std::string host = hostport.substr(0, colon);
  int port = std::stoi(hostport.substr(colon + 1));
  sockaddr_in sa;
  std::memset(&sa, 0, sizeof(sa));
  sa.sin_family = AF_INET;
  sa.sin_port = htons(static_cast<uint16_t>(port));
  TMA_CHECK(inet_pton(AF_INET, host.c_str(), &sa.sin_addr) == 1,
            "inet_pton failed for " + host);
  return sa;
}

int tcp_timeout_ms() {
  static int t = [] {
    const char* e = std::getenv("TORCHMPI_TCP_TIMEOUT_MS");
    return e ? std::atoi(e) : 600000;  // generous: ranks can skew minutes
                                       // during MIOpen warmup/find phases
  }();
  return t;
}

}  // namespace

TcpMesh::TcpMesh(const std::string& bind_addr) {
  listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
  TMA_CHECK(listen_fd_ >= 0, "socket() failed");
  int one = 1;
  setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  sockaddr_in sa;
  std::memset(&sa, 0, sizeof(sa));
  sa.sin_family = AF_INET;
  sa.sin_port = 0;  // ephemeral
  TMA_CHECK(inet_pton(AF_INET, bind_addr.c_str(), &sa.sin_addr) == 1,
            "inet_pton failed for bind addr " + bind_addr);
  TMA_CHECK(bind(listen_fd_, reinterpret_cast<sockaddr*>(&sa), sizeof(sa)) == 0,
            "bind failed");
  TMA_CHECK(listen(listen_fd_, 128) == 0, "listen failed");
  socklen_t len = sizeof(sa);
  TMA_CHECK(getsockname(listen_fd_, reinterpret_cast<sockaddr*>(&sa), &len) == 0,
            "getsockname failed");
  port_ = ntohs(sa.sin_port);
}

TcpMesh::~TcpMesh() {
  for (int fd : fds_)
    if (fd >= 0) close(fd);
  if (listen_fd_ >= 0) close(listen_fd_);
}

void TcpMesh::connect_mesh(int rank, const std::vector<std::string>& addrs) {
  rank_ = rank;
  int world = static_cast<int>(addrs.size());
  fds_.assign(world, -1);

  // Connect to every lower rank (blocking connect with retry: the peer's
  // listener exists — its address was published — but accept backlog races
  // are possible under heavy spawn).
  for (int peer = 0; peer < rank; ++peer) {
    sockaddr_in sa = parse_addr(addrs[peer]);
    int fd = -1;
    for (int attempt = 0; attempt < 200; ++attempt) {
      fd = socket(AF_INET, SOCK_STREAM, 0);
      TMA_CHECK(fd >= 0, "socket() failed");
      if (connect(fd, reinterpret_cast<sockaddr*>(&sa), sizeof(sa)) == 0) break;
      close(fd);
      fd = -1;
      usleep(50 * 1000);
    }
    TMA_CHECK(fd >= 0, "connect to rank " + std::to_string(peer) + " failed");
    set_sockopts(fd);
    uint32_t me = static_cast<uint32_t>(rank);
    TMA_CHECK(::send(fd, &me, 4, 0) == 4, "handshake send failed");
    set_nonblocking(fd);
    fds_[peer] = fd;
  }

  // Accept one connection from every higher rank.
  for (int i = rank + 1; i < world; ++i) {
    int fd;
    do {
      fd = accept(listen_fd_, nullptr, nullptr);
    } while (fd < 0 && errno == EINTR);
    TMA_CHECK(fd >= 0, "accept failed");
    set_sockopts(fd);
    uint32_t peer = 0;
    ssize_t got = 0;
    while (got < 4) {
      ssize_t r = ::recv(fd, reinterpret_cast<char*>(&peer) + got, 4 - got, 0);
      TMA_CHECK(r > 0, "handshake recv failed");
      got += r;
    }
    TMA_CHECK(static_cast<int>(peer) > rank && static_cast<int>(peer) < world,
              "bad handshake rank");
    set_nonblocking(fd);
    fds_[peer] = fd;
  }
}

void TcpMesh::send_all(int peer, const void* buf, size_t n) {
  duplex(peer, buf, n, -1, nullptr, 0);
}

void TcpMesh::recv_all(int peer, void* buf, size_t n) {
  duplex(-1, nullptr, 0, peer, buf, n);
}

void TcpMesh::duplex(int speer, const void* sbuf, size_t sn, int rpeer, void* rbuf,
                     size_t rn) {
  TMA_CHECK(speer != rank_ && rpeer != rank_, "self send/recv not supported here");
  int sfd = speer >= 0 ? fds_[speer] : -1;
  int rfd = rpeer >= 0 ? fds_[rpeer] : -1;
  TMA_CHECK(speer < 0 || sfd >= 0, "no connection to send peer");
  TMA_CHECK(rpeer < 0 || rfd >= 0, "no connection to recv peer");

  const char* sp = static_cast<const char*>(sbuf);
  char* rp = static_cast<char*>(rbuf);
  size_t sent = 0, rcvd = 0;

  while (sent < sn || rcvd < rn) {
    // Try to make progress on both directions; poll when neither can move.
    bool progressed = false;
    if (sent < sn) {
      ssize_t r = ::send(sfd, sp + sent, sn - sent, MSG_NOSIGNAL);
      if (r > 0) {
        sent += static_cast<size_t>(r);
        progressed = true;
      } else if (r < 0 && errno != EAGAIN && errno != EWOULDBLOCK &&
                 errno != EINTR) {
        throw std::runtime_error("torchmpi_amd: tcp send failed: " +
                                 std::string(strerror(errno)));
      }
    }
    if (rcvd < rn) {
      ssize_t r = ::recv(rfd, rp + rcvd, rn - rcvd, 0);
      if (r > 0) {
        rcvd += static_cast<size_t>(r);
        progressed = true;
      } else if (r == 0) {
        throw std::runtime_error("torchmpi_amd: peer closed connection");
      } else if (errno != EAGAIN && errno != EWOULDBLOCK && errno != EINTR) {
        throw std::runtime_error("torchmpi_amd: tcp recv failed: " +
                                 std::string(strerror(errno)));
      }
    }
    if (!progressed) {
      // Block until something is ready.
      pollfd ps[2];
      int np = 0;
      if (sent < sn) ps[np++] = {sfd, POLLOUT, 0};
      if (rcvd < rn) ps[np++] = {rfd, POLLIN, 0};
      int r = poll(ps, np, tcp_timeout_ms());
      if (r < 0 && errno == EINTR) continue;  // signal during poll: retry
      TMA_CHECK(r > 0, "tcp duplex poll timeout (deadlock or dead peer?)");
    }
  }
}

}  // namespace tma
