#include "comm.h"

#include <arpa/inet.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <poll.h>
#include <unistd.h>

#include <chrono>
#include <cstring>
#include <stdexcept>
#include <thread>

namespace hvd {

namespace {

void set_nodelay(int fd) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}

// Mesh sockets MUST be non-blocking: on Linux a blocking send() on a
// stream socket queues the ENTIRE request in-kernel before returning, so
// a poll loop that calls send() for the full remainder stops servicing
// POLLIN for the duration — two ranks exchanging buffers larger than the
// socket queues then deadlock with both tx and rx windows full (observed
// with ~32 MB ring segments; root cause of the round-2 mesh hang).
void set_nonblock(int fd) {
  int flags = fcntl(fd, F_GETFL, 0);
  if (flags >= 0) fcntl(fd, F_SETFL, flags | O_NONBLOCK);
}

[[noreturn]] void comm_error(const std::string& what) {
  throw std::runtime_error("horovod_amd comm: " + what + " (" +
                           std::strerror(errno) + ")");
}

}  // namespace

StarComm::~StarComm() { Shutdown(); }

void StarComm::Init(int rank, int size, const std::string& addr, int port,
                    double timeout_sec) {
  rank_ = rank;
  size_ = size;
  if (size_ <= 1) {
    alive_ = true;
    return;
  }
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::duration<double>(timeout_sec);
  if (rank_ == 0) {
    fds_.assign(size_, -1);
    listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) comm_error("socket");
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in sa{};
    sa.sin_family = AF_INET;
    sa.sin_addr.s_addr = INADDR_ANY;
    sa.sin_port = htons((uint16_t)port);
    if (bind(listen_fd_, (sockaddr*)&sa, sizeof(sa)) != 0) comm_error("bind");
    if (listen(listen_fd_, size_) != 0) comm_error("listen");
    int connected = 0;
    while (connected < size_ - 1) {
      // bounded accept: poll with the remaining init deadline so a missing
      // worker turns into an error instead of an eternal hang
      auto remain = std::chrono::duration<double>(
                        deadline - std::chrono::steady_clock::now())
                        .count();
      if (remain <= 0) comm_error("accept timeout (worker never connected)");
      struct pollfd pfd{listen_fd_, POLLIN, 0};
      int pr = poll(&pfd, 1, (int)(remain * 1000));
      if (pr == 0) comm_error("accept timeout (worker never connected)");
      if (pr < 0) comm_error("poll");
      int fd = accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) comm_error("accept");
      set_nodelay(fd);
      int32_t peer_rank = -1;
      RecvRaw(fd, &peer_rank, sizeof(peer_rank));
      if (peer_rank <= 0 || peer_rank >= size_) comm_error("bad peer rank");
      fds_[peer_rank] = fd;
      ++connected;
    }
  } else {
    fds_.assign(1, -1);
    // resolve
    addrinfo hints{}, *res = nullptr;
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    std::string port_s = std::to_string(port);
    if (getaddrinfo(addr.c_str(), port_s.c_str(), &hints, &res) != 0 || !res)
      comm_error("getaddrinfo " + addr);
    int fd = -1;
    while (true) {
      fd = socket(AF_INET, SOCK_STREAM, 0);
      if (fd < 0) comm_error("socket");
      if (connect(fd, res->ai_addr, res->ai_addrlen) == 0) break;
      close(fd);
      fd = -1;
      if (std::chrono::steady_clock::now() > deadline) {
        freeaddrinfo(res);
        comm_error("connect timeout to " + addr + ":" + port_s);
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(50));
    }
    freeaddrinfo(res);
    set_nodelay(fd);
    int32_t my_rank = rank_;
    SendRaw(fd, &my_rank, sizeof(my_rank));
    fds_[0] = fd;
  }
  alive_ = true;
}

void StarComm::Shutdown() {
  for (int fd : fds_)
    if (fd >= 0) close(fd);
  fds_.clear();
  if (listen_fd_ >= 0) close(listen_fd_);
  listen_fd_ = -1;
  alive_ = false;
}

void StarComm::SendRaw(int fd, const void* data, size_t len) {
  const char* p = (const char*)data;
  while (len > 0) {
    ssize_t n = ::send(fd, p, len, MSG_NOSIGNAL);
    if (n <= 0) comm_error("send");
    p += n;
    len -= (size_t)n;
  }
}

void StarComm::RecvRaw(int fd, void* data, size_t len) {
  char* p = (char*)data;
  while (len > 0) {
    ssize_t n = ::recv(fd, p, len, 0);
    if (n <= 0) comm_error("recv");
    p += n;
    len -= (size_t)n;
  }
}

void StarComm::SendFrame(int fd, const std::string& payload) {
  uint32_t len = (uint32_t)payload.size();
  SendRaw(fd, &len, sizeof(len));
  if (len) SendRaw(fd, payload.data(), len);
}

std::string StarComm::RecvFrame(int fd) {
  uint32_t len = 0;
  RecvRaw(fd, &len, sizeof(len));
  std::string s(len, '\0');
  if (len) RecvRaw(fd, &s[0], len);
  return s;
}

std::vector<std::string> StarComm::Gather(const std::string& payload) {
  if (size_ == 1) return {payload};
  if (rank_ == 0) {
    std::vector<std::string> out(size_);
    out[0] = payload;
    for (int r = 1; r < size_; ++r) out[r] = RecvFrame(fds_[r]);
    return out;
  }
  SendFrame(fds_[0], payload);
  return {};
}

std::string StarComm::Bcast(const std::string& payload) {
  if (size_ == 1) return payload;
  if (rank_ == 0) {
    for (int r = 1; r < size_; ++r) SendFrame(fds_[r], payload);
    return payload;
  }
  return RecvFrame(fds_[0]);
}

void StarComm::Barrier() {
  Gather("");
  Bcast("");
}

void StarComm::BitAnd(std::string& bits, size_t verify_tail) {
  auto all = Gather(bits);
  if (rank_ == 0) {
    std::string acc = all[0];
    size_t head = acc.size() - std::min(verify_tail, acc.size());
    for (int r = 1; r < size_; ++r) {
      if (all[r].size() != acc.size())
        throw std::runtime_error("horovod_amd comm: bitvector length mismatch");
      for (size_t i = 0; i < head; ++i) acc[i] &= all[r][i];
      for (size_t i = head; i < acc.size(); ++i) {
        if (all[r][i] != acc[i])
          throw std::runtime_error(
              "horovod_amd: response-stream divergence detected — rank " +
              std::to_string(r) + "'s executed-response hash differs from "
              "rank 0's (fast-path cache state desynchronized; this would "
              "deadlock inside RCCL). Set HOROVOD_CACHE_CAPACITY=0 or "
              "report a bug.");
      }
    }
    bits = Bcast(acc);
  } else {
    bits = Bcast("");
  }
}

void StarComm::BitOr(std::string& bits) {
  auto all = Gather(bits);
  if (rank_ == 0) {
    std::string acc = all[0];
    for (int r = 1; r < size_; ++r) {
      if (all[r].size() != acc.size())
        throw std::runtime_error("horovod_amd comm: bitvector length mismatch");
      for (size_t i = 0; i < acc.size(); ++i) acc[i] |= all[r][i];
    }
    bits = Bcast(acc);
  } else {
    bits = Bcast("");
  }
}

std::string StarComm::ScatterFrames(const std::vector<std::string>& frames) {
  if (size_ == 1) return frames.empty() ? std::string() : frames[0];
  if (rank_ == 0) {
    for (int r = 1; r < size_; ++r)
      SendFrame(fds_[r], r < (int)frames.size() ? frames[r] : std::string());
    return frames.empty() ? std::string() : frames[0];
  }
  return RecvFrame(fds_[0]);
}


// ---------------------------------------------------------------------------
// MeshComm
// ---------------------------------------------------------------------------

MeshComm::~MeshComm() { Shutdown(); }

int MeshComm::fd_of(int peer) const {
  int fd = fds_[peer];
  if (fd < 0) comm_error("mesh: no link to rank " + std::to_string(peer));
  return fd;
}

void MeshComm::Init(StarComm& star, const std::string& root_addr,
                    double timeout_sec) {
  rank_ = star.rank();
  size_ = star.size();
  if (size_ <= 1) {
    alive_ = true;
    return;
  }
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::duration<double>(timeout_sec);
  fds_.assign(size_, -1);

  // 1. listener on an ephemeral port
  listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
  if (listen_fd_ < 0) comm_error("mesh socket");
  int one = 1;
  setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  sockaddr_in sa{};
  sa.sin_family = AF_INET;
  sa.sin_addr.s_addr = INADDR_ANY;
  sa.sin_port = 0;
  if (bind(listen_fd_, (sockaddr*)&sa, sizeof(sa)) != 0)
    comm_error("mesh bind");
  if (listen(listen_fd_, size_) != 0) comm_error("mesh listen");
  socklen_t slen = sizeof(sa);
  getsockname(listen_fd_, (sockaddr*)&sa, &slen);
  int my_port = ntohs(sa.sin_port);

  // 2. my address as peers can reach it: rank 0 advertises the address the
  // workers already used for the star; a worker advertises the local IP of
  // its star socket (the interface that routes to the job).
  std::string my_ip;
  if (rank_ == 0) {
    my_ip = root_addr;
  } else {
    sockaddr_in la{};
    socklen_t ll = sizeof(la);
    getsockname(star.fds_[0], (sockaddr*)&la, &ll);
    char buf[64];
    inet_ntop(AF_INET, &la.sin_addr, buf, sizeof(buf));
    my_ip = buf;
  }
  std::string endpoint = my_ip + ":" + std::to_string(my_port);
  auto eps = star.Gather(endpoint);
  {
    std::string joined;
    if (star.is_root())
      for (auto& e : eps) joined += e + "\n";
    joined = star.Bcast(joined);
    eps.clear();
    size_t pos = 0;
    while (pos < joined.size()) {
      size_t nl = joined.find('\n', pos);
      eps.push_back(joined.substr(pos, nl - pos));
      pos = nl + 1;
    }
  }

  // 3. connect to every LOWER rank (their listeners already exist: the
  // endpoint exchange above is a barrier), then accept the higher ranks.
  for (int peer = 0; peer < rank_; ++peer) {
    auto colon = eps[peer].rfind(':');
    std::string ip = eps[peer].substr(0, colon);
    std::string port_s = eps[peer].substr(colon + 1);
    addrinfo hints{}, *res = nullptr;
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    if (getaddrinfo(ip.c_str(), port_s.c_str(), &hints, &res) != 0 || !res)
      comm_error("mesh getaddrinfo " + eps[peer]);
    int fd = -1;
    while (true) {
      fd = socket(AF_INET, SOCK_STREAM, 0);
      if (fd < 0) comm_error("mesh socket");
      if (connect(fd, res->ai_addr, res->ai_addrlen) == 0) break;
      close(fd);
      if (std::chrono::steady_clock::now() > deadline) {
        freeaddrinfo(res);
        comm_error("mesh connect timeout to " + eps[peer]);
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(20));
    }
    freeaddrinfo(res);
    set_nodelay(fd);
    int32_t me = rank_;
    const char* pb = (const char*)&me;
    size_t left = sizeof(me);
    while (left) {
      ssize_t n = ::send(fd, pb, left, MSG_NOSIGNAL);
      if (n <= 0) comm_error("mesh handshake send");
      pb += n;
      left -= (size_t)n;
    }
    fds_[peer] = fd;
    set_nonblock(fd);
  }
  int expected = size_ - 1 - rank_;
  while (expected > 0) {
    auto remain = std::chrono::duration<double>(
                      deadline - std::chrono::steady_clock::now())
                      .count();
    if (remain <= 0) comm_error("mesh accept timeout");
    struct pollfd pfd{listen_fd_, POLLIN, 0};
    int pr = poll(&pfd, 1, (int)(remain * 1000));
    if (pr <= 0) comm_error("mesh accept timeout/poll");
    int fd = accept(listen_fd_, nullptr, nullptr);
    if (fd < 0) comm_error("mesh accept");
    set_nodelay(fd);
    int32_t peer = -1;
    char* pb = (char*)&peer;
    size_t left = sizeof(peer);
    while (left) {
      ssize_t n = ::recv(fd, pb, left, 0);
      if (n <= 0) comm_error("mesh handshake recv");
      pb += n;
      left -= (size_t)n;
    }
    if (peer <= rank_ || peer >= size_) comm_error("mesh bad peer rank");
    fds_[peer] = fd;
    set_nonblock(fd);
    --expected;
  }
  close(listen_fd_);
  listen_fd_ = -1;
  alive_ = true;
}

void MeshComm::Shutdown() {
  for (int fd : fds_)
    if (fd >= 0) close(fd);
  fds_.clear();
  if (listen_fd_ >= 0) close(listen_fd_);
  listen_fd_ = -1;
  alive_ = false;
}

void MeshComm::Send(int peer, const void* data, size_t len) {
  SendRecv2(peer, data, len, -1, nullptr, 0);
}

void MeshComm::Recv(int peer, void* data, size_t len) {
  SendRecv2(-1, nullptr, 0, peer, data, len);
}

void MeshComm::SendRecv2(int send_peer, const void* out, size_t out_len,
                         int recv_peer, void* in, size_t in_len) {
  // self-exchange degenerates to memcpy
  if (send_peer == rank_ && recv_peer == rank_) {
    if (in_len != out_len) comm_error("mesh self-exchange length mismatch");
    if (in_len) std::memcpy(in, out, in_len);
    return;
  }
  if (send_peer == rank_ || recv_peer == rank_)
    comm_error("mesh asymmetric self-exchange unsupported");
  size_t so = 0, ro = 0;
  int sfd = (send_peer >= 0 && out_len) ? fd_of(send_peer) : -1;
  int rfd = (recv_peer >= 0 && in_len) ? fd_of(recv_peer) : -1;
  while ((sfd >= 0 && so < out_len) || (rfd >= 0 && ro < in_len)) {
    struct pollfd p[2];
    int n = 0, si = -1, ri = -1;
    if (sfd >= 0 && so < out_len) {
      p[n] = {sfd, POLLOUT, 0};
      si = n++;
    }
    if (rfd >= 0 && ro < in_len) {
      p[n] = {rfd, POLLIN, 0};
      ri = n++;
    }
    int pr = poll(p, n, 300000);  // 5-minute safety: a dead peer -> error
    if (pr == 0) comm_error("mesh transfer timeout");
    if (pr < 0) comm_error("mesh poll");
    if (si >= 0 && (p[si].revents & (POLLOUT | POLLERR | POLLHUP))) {
      ssize_t w = ::send(sfd, (const char*)out + so, out_len - so,
                         MSG_NOSIGNAL);
      if (w < 0 && errno != EAGAIN && errno != EWOULDBLOCK)
        comm_error("mesh send");
      if (w > 0) so += (size_t)w;
    }
    if (ri >= 0 && (p[ri].revents & (POLLIN | POLLERR | POLLHUP))) {
      ssize_t r = ::recv(rfd, (char*)in + ro, in_len - ro, 0);
      if (r == 0) comm_error("mesh recv: peer closed");
      if (r < 0 && errno != EAGAIN && errno != EWOULDBLOCK)
        comm_error("mesh recv (peer died?)");
      if (r > 0) ro += (size_t)r;
    }
  }
}

}  // namespace hvd
