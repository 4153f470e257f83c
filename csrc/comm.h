// TCP star control channel.
//
// MI355X-native replacement for the reference's control plane (MPI or Gloo
// controllers, horovod/common/mpi/mpi_controller.cc and
// common/gloo/gloo_controller.cc).  On a single 8-GPU MI355X node (and small
// clusters) the negotiation traffic is tiny (~100 B/cycle bitvectors), so a
// hand-rolled star topology over TCP — rank 0 listens, everyone connects —
// replaces the MPI/Gloo dependency entirely.  All collectives here are
// lock-step: every rank calls them in the same order, so the framing needs
// no tags.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace hvd {

class StarComm {
 public:
  StarComm() = default;
  ~StarComm();

  // Connect the mesh.  addr/port locate rank 0's listener.  Safe to call
  // with size == 1 (no sockets at all).
  void Init(int rank, int size, const std::string& addr, int port,
            double timeout_sec = 120.0);
  void Shutdown();

  int rank() const { return rank_; }
  int size() const { return size_; }
  bool is_root() const { return rank_ == 0; }

  // Collectives (lock-step; every rank participates).
  // Gather: returns size_ payloads at root (index = rank), empty elsewhere.
  std::vector<std::string> Gather(const std::string& payload);
  // Bcast: root's payload is returned on every rank.
  std::string Bcast(const std::string& payload);
  void Barrier();
  // Byte-wise AND / OR across ranks (cache-bitvector sync; reference:
  // mpi_controller.cc CrossRankBitwiseAnd/Or).
  void BitAnd(std::string& bits);
  void BitOr(std::string& bits);
  // Root sends frames[r] to each rank r; every rank returns its own frame.
  std::string ScatterFrames(const std::vector<std::string>& frames);

  bool alive() const { return alive_; }

 private:
  void SendFrame(int fd, const std::string& payload);
  std::string RecvFrame(int fd);
  void SendRaw(int fd, const void* data, size_t len);
  void RecvRaw(int fd, void* data, size_t len);

  int rank_ = 0;
  int size_ = 1;
  bool alive_ = false;
  int listen_fd_ = -1;
  // root: fds_[r] = connection to rank r (fds_[0] unused);
  // worker: fds_[0] = connection to root.
  std::vector<int> fds_;
};

}  // namespace hvd
