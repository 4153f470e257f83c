// horovod_amd core runtime: global state, tensor queue, background cycle
// loop, enqueue API and CPU data plane.
//
// Re-design of the reference's horovod/common/operations.{cc,h} (N1),
// tensor_queue.{cc,h} (N13) and the CPU side of the op set for an MI355X
// single-node-first deployment: PyTorch-only tensors, a TCP star control
// plane (comm.h) and RCCL-over-xGMI as the only GPU data plane (gpu.h).
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
#include "comm.h"
#include "controller.h"
#include "message.h"

namespace hvd {

// ---------------------------------------------------------------------------
// TensorQueue (reference: common/tensor_queue.{cc,h})
// ---------------------------------------------------------------------------
class TensorQueue {
 public:
  Status Add(Request req, TensorTableEntry entry);
  // Atomic multi-insert (grouped ops): all-or-nothing on duplicate names.
  Status AddMulti(std::vector<Request>& reqs,
                  std::vector<TensorTableEntry>& entries);
  std::vector<Request> PopMessages();
  bool PopEntry(int32_t set_id, const std::string& name, TensorTableEntry& out);
  void FailAll(const Status& s);
  size_t size() const;
  bool has_messages() const;

 private:
  mutable std::mutex mu_;
  std::condition_variable cv_;
  std::deque<Request> messages_;
  std::unordered_map<std::string, TensorTableEntry> table_;
};

// ---------------------------------------------------------------------------
// HandleManager (reference: horovod/torch/handle_manager.{cc,h})
// ---------------------------------------------------------------------------
struct HandleState {
  std::mutex mu;
  std::condition_variable cv;
  bool done = false;
  Status status;
  std::vector<at::Tensor> outputs;
  at::Tensor extra;         // alltoall: received_splits
  int32_t result_int = -1;  // join: last joined rank
};

class HandleManager {
 public:
  int Allocate(int n_outputs);
  std::shared_ptr<HandleState> Get(int handle);
  bool Poll(int handle);
  // Blocks (caller must NOT hold the GIL).
  Status Wait(int handle, std::vector<at::Tensor>& outputs, at::Tensor& extra,
              int32_t* result_int);
  void MarkDone(int handle, const Status& s);
  void FailAll(const Status& s);
  void Drop(int handle);

 private:
  std::mutex mu_;
  int next_ = 0;
  std::unordered_map<int, std::shared_ptr<HandleState>> handles_;
};

// ---------------------------------------------------------------------------
// Global state + API
// ---------------------------------------------------------------------------
struct GlobalState {
  std::atomic<bool> initialized{false};
  std::atomic<bool> shutting_down{false};
  std::atomic<bool> shutdown_requested{false};
  std::atomic<bool> aborted{false};
  std::string abort_reason;

  int rank = 0, size = 1, local_rank = 0, local_size = 1, cross_rank = 0,
      cross_size = 1;

  StarComm comm;   // control plane (negotiation, bitvectors, bootstrap)
  MeshComm mesh;   // CPU data plane (direct member-to-member links)
  std::unique_ptr<Controller> controller;
  TensorQueue queue;
  HandleManager handles;
  std::thread bg_thread;

  // cycle pacing: cycles are paced at cycle_time so gradient bursts fuse
  // into large buckets; flush() (called from synchronize()) cuts the wait
  // so the trailing bucket of a step fires immediately.
  std::mutex pace_mu;
  std::condition_variable pace_cv;
  bool flush_requested = false;

  // join(): this rank's joined state + zero-substitute device, per set.
  std::mutex join_mu;
  std::unordered_map<int32_t, bool> local_joined;
  std::unordered_map<int32_t, int> join_device;

  // hot-swappable at runtime (hvd.start_timeline): accessed via atomic
  // shared_ptr loads from the background thread.
  std::shared_ptr<class Timeline> timeline;
};

GlobalState& State();

std::shared_ptr<class Timeline> GetTimeline(GlobalState& st);
void SetTimeline(GlobalState& st, std::shared_ptr<class Timeline> t);

// Lifecycle -----------------------------------------------------------------
void InitHorovod(int rank, int size, int local_rank, int local_size,
                 int cross_rank, int cross_size, const std::string& addr,
                 int port, const ControllerConfig& cfg);
void ShutdownHorovod();
// Fail all in-flight + future collectives with ABORTED (elastic interrupt).
void InterruptHorovod(const std::string& why);
bool IsInitialized();
// Cut the current cycle-pacing wait (the caller is about to block on
// results): lets the background thread negotiate the tail immediately.
void FlushCycle();

// Enqueue API (reference: EnqueueTensor* operations.cc:1408-2057) -----------
// wire_dtype may differ from the tensors' dtype: the fusion pack kernel
// converts on the fly (fp16/bf16 gradient compression with zero extra
// memory passes — the MI355X replacement for the reference's python-side
// Compression.fp16).
int EnqueueAllreduceMulti(std::vector<at::Tensor> tensors,
                          std::vector<at::Tensor> outputs,
                          std::vector<std::string> names, ReduceOp op,
                          double prescale, double postscale,
                          int32_t process_set_id, DataType wire_dtype);
int EnqueueAllgather(at::Tensor tensor, const std::string& name,
                     int32_t process_set_id);
int EnqueueBroadcast(at::Tensor tensor, at::Tensor output, int root_rank,
                     const std::string& name, int32_t process_set_id);
int EnqueueAlltoall(at::Tensor tensor, at::Tensor splits, const std::string& name,
                    int32_t process_set_id);
int EnqueueReducescatter(at::Tensor tensor, const std::string& name, ReduceOp op,
                         double prescale, double postscale, int32_t process_set_id);
int EnqueueJoin(int device, int32_t process_set_id);
int EnqueueBarrier(int32_t process_set_id);

}  // namespace hvd
