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
  // mpi_controller.cc CrossRankBitwiseAnd/Or).  When verify_tail > 0 the
  // last verify_tail bytes of the frame are NOT reduced: the root asserts
  // they are identical on every rank and throws on divergence (the
  // response-stream hash cross-check — one divergent fast-path cache state
  // would otherwise deadlock inside RCCL with no diagnosis).
  void BitAnd(std::string& bits, size_t verify_tail = 0);
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

  friend class MeshComm;  // port exchange rides the star at init
};

// Full-mesh TCP data plane: direct member-to-member links for the CPU
// collectives (ring allreduce, pairwise-exchange allgather/alltoall,
// binomial broadcast), replacing the star gather+bcast that moved 2n x
// data through rank 0 (round-1 weakness; reference analogue:
// mpi_operations.cc:83-545 real MPI collectives among members).
//
// Only the background thread touches mesh sockets, in broadcast-ordered
// response order, so per-pair framing needs no tags.  All transfers are
// poll-driven full-duplex (SendRecv2): a ring step where every rank sends
// and receives concurrently cannot deadlock on TCP buffer limits.
class MeshComm {
 public:
  MeshComm() = default;
  ~MeshComm();

  // Collective over the (already-initialized) star: every global rank must
  // call this together.  root_addr is rank 0's address as workers know it.
  void Init(StarComm& star, const std::string& root_addr,
            double timeout_sec = 120.0);
  void Shutdown();
  bool alive() const { return alive_; }

  void Send(int peer, const void* data, size_t len);
  void Recv(int peer, void* data, size_t len);
  // Full-duplex: send `out` to send_peer while receiving `in` from
  // recv_peer (they may be the same peer, or self => memcpy).
  void SendRecv2(int send_peer, const void* out, size_t out_len,
                 int recv_peer, void* in, size_t in_len);

 private:
  int fd_of(int peer) const;

  int rank_ = 0;
  int size_ = 1;
  bool alive_ = false;
  int listen_fd_ = -1;
  std::vector<int> fds_;  // fds_[r] = direct link to rank r (self = -1)
};

}  // namespace hvd
