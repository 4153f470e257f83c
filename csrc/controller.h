// Negotiation controller.
//
// Re-design of the reference's Controller/ResponseCache/StallInspector
// (horovod/common/controller.cc:74-1139, common/response_cache.{cc,h},
// common/stall_inspector.{cc,h}) for a TCP star control plane.
//
// Differences from the reference, by design:
//  * ONE global controller negotiates every process set (the reference builds
//    a controller per set).  The coordinator is global rank 0; responses are
//    broadcast to all ranks and filtered by set membership at execution time.
//  * The cache bitvector carries two control flags (shutdown via AND round,
//    needs-slow-path via OR round) instead of a separate state struct.
//
// Protocol per cycle (all ranks, lock-step):
//  1. classify new requests: cache hit -> wait on bitvector; miss -> send to
//     coordinator once.
//  2. round A: bitwise AND of [hit bits | shutdown flag];
//     round B: bitwise OR of [invalid bits | slowpath flag | join flag].
//  3. common hits (AND minus OR-invalid) become responses straight from the
//     cache on every rank — the steady-state training path never touches the
//     coordinator (reference fast path: controller.cc:195-252).
//  4. if any rank flagged slow path: Gather(RequestList) -> coordinator
//     counts, constructs + fuses responses -> Bcast(ResponseList).
#pragma once

#include <chrono>
#include <deque>
#include <list>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <vector>

#include "autotuner.h"
#include "comm.h"
#include "common.h"
#include "message.h"

namespace hvd {

struct ProcessSetInfo {
  int32_t id = 0;
  std::vector<int32_t> ranks;          // global ranks, sorted
  std::unordered_set<int32_t> joined;  // ranks that called join()
  int32_t last_joined_rank = -1;

  bool contains(int32_t rank) const {
    for (auto r : ranks)
      if (r == rank) return true;
    return false;
  }
  int32_t local_index(int32_t rank) const {
    for (size_t i = 0; i < ranks.size(); ++i)
      if (ranks[i] == rank) return (int32_t)i;
    return -1;
  }
};

// LRU response cache with rank-consistent slot numbering (slots mutate only
// on broadcast responses, which every rank sees in the same order).
class ResponseCache {
 public:
  enum class State { MISS, HIT, INVALID };

  void set_capacity(size_t cap) { capacity_ = cap; }
  size_t capacity() const { return capacity_; }

  State Lookup(const Request& req) const;
  int SlotOf(const std::string& name) const;
  const Response& Get(int slot) const;
  const Request& GetRequest(int slot) const;
  // Insert/update from an executed (possibly fused) response; deterministic
  // across ranks.  Single-tensor granularity.  my_local_index: this rank's
  // index in the response's process set (alltoall split-row extraction).
  void Put(const Response& response, const std::vector<Request>& reqs,
           int32_t my_local_index = -1);
  void Evict(int slot);
  void EvictSet(int32_t process_set_id);
  // Evict everything (cache-off TUNE: the reference clears on disable so a
  // later re-enable rebuilds every rank's state from scratch — stale live
  // slots would otherwise classify HIT on one rank while a peer's copy is
  // already in flight on the slow path: cross-rank deadlock).
  void Clear();
  size_t num_slots() const { return entries_.size(); }
  bool slot_live(int slot) const { return entries_[slot].live; }

 private:
  struct Entry {
    Request request;   // signature for hit checking
    Response response; // single-tensor response
    bool live = false;
    uint64_t last_use = 0;
  };
  size_t capacity_ = 1024;
  uint64_t use_tick_ = 0;
  std::vector<Entry> entries_;              // slot -> entry
  std::unordered_map<std::string, int> name_to_slot_;
  std::vector<int> free_slots_;

  int AllocSlot();
};

// Coordinator-side accounting of which ranks announced which tensors
// (reference: controller.cc IncrementTensorCount / message table).
struct PendingTensor {
  std::vector<Request> requests;       // one per announcing rank
  std::vector<bool> have;              // by set-local index
  int count = 0;
  std::chrono::steady_clock::time_point first_seen;
};

struct ControllerConfig {
  int64_t fusion_threshold_bytes = 64ll << 20;
  double cycle_time_ms = 1.0;
  size_t cache_capacity = 1024;
  double stall_warning_sec = 60.0;
  double stall_shutdown_sec = 0.0;  // 0 = never
  bool timeline_enabled = false;
};

class Controller {
 public:
  Controller(StarComm* comm, int rank, int size, ControllerConfig cfg);

  // Run one negotiation cycle.  new_requests: popped from the tensor queue
  // this cycle.  shutdown_requested: this rank wants to shut down.
  // Returns the ordered list of responses every rank must execute, plus
  // whether global shutdown was agreed.
  ResponseList RunCycle(std::vector<Request> new_requests,
                        bool shutdown_requested);

  ProcessSetInfo& process_set(int32_t id) { return process_sets_.at(id); }
  bool has_process_set(int32_t id) const { return process_sets_.count(id) > 0; }
  int32_t AddProcessSet(const std::vector<int32_t>& ranks);
  void RemoveProcessSet(int32_t id);
  const std::unordered_map<int32_t, ProcessSetInfo>& process_sets() const {
    return process_sets_;
  }

  void set_fusion_threshold(int64_t bytes) { cfg_.fusion_threshold_bytes = bytes; }
  int64_t fusion_threshold() const { return cfg_.fusion_threshold_bytes; }
  void set_cycle_time_ms(double ms) { cfg_.cycle_time_ms = ms; }
  double cycle_time_ms() const { return cfg_.cycle_time_ms; }
  ControllerConfig& config() { return cfg_; }
  // tensors still negotiating (cache-hit bits not yet agreed, or requests at
  // the coordinator) — the cycle loop must not sleep while any exist
  bool has_pending() const { return !cached_pending_.empty() || !inflight_.empty(); }
  void ApplyTune(int64_t fusion_bytes, double cycle_time_ms,
                 bool cache_enabled = true) {
    cfg_.fusion_threshold_bytes = fusion_bytes;
    cfg_.cycle_time_ms = cycle_time_ms;
    if (cache_enabled_ && !cache_enabled) cache_.Clear();
    cache_enabled_ = cache_enabled;
  }

 private:
  // Slow path, coordinator side.
  std::vector<Response> CoordinatorProcess(
      const std::vector<std::string>& gathered);
  Response ConstructResponse(const std::string& name, PendingTensor& pt);
  std::vector<Response> FuseResponses(std::deque<Response>& queue);
  void CheckForStalledTensors();

  StarComm* comm_;
  int rank_;
  int size_;
  ControllerConfig cfg_;
  ResponseCache cache_;

  // Requests waiting on the cache bitvector (name -> request, FIFO).
  std::list<Request> cached_pending_;
  // Requests sent to the coordinator, awaiting a response.
  std::unordered_set<std::string> inflight_;

  // coordinator state
  std::unordered_map<std::string, PendingTensor> table_;
  std::deque<std::string> arrival_order_;  // table keys, first-seen order
  std::deque<Response> ready_responses_;  // completed, awaiting fusion window
  // grouped-op holding area: set:group_key -> completed singles
  std::unordered_map<std::string, std::vector<Response>> group_hold_;
  std::unordered_set<std::string> poisoned_groups_;

  std::unordered_map<int32_t, ProcessSetInfo> process_sets_;
  int32_t next_set_id_ = 1;
  std::chrono::steady_clock::time_point last_stall_check_;
  std::chrono::steady_clock::time_point start_time_;

  // autotuner (rank 0 only; results propagate via TUNE responses)
  std::unique_ptr<Autotuner> autotuner_;
  bool pending_tune_ = false;
  // autotuner categorical arm: when off, every request renegotiates (the
  // reference's cache on/off boolean); toggled rank-synchronously via TUNE
  bool cache_enabled_ = true;
  // HOROVOD_CHECK_RESPONSE_STREAM=1: running FNV-1a hash of every executed
  // response, cross-checked in the bitvector round each cycle — catches a
  // diverging fast path before it deadlocks RCCL
  bool check_stream_ = false;
  uint64_t response_hash_ = 1469598103934665603ull;
  // process sets this rank has JOINed (votes all cache slots until the
  // set-wide JOIN response arrives; reference controller.cc:130-134)
  std::unordered_set<int32_t> my_joined_;
  // HOROVOD_TRACE_CYCLES=<prefix>: per-cycle debug log (prefix.<rank>)
  std::FILE* trace_ = nullptr;
  uint64_t trace_cycle_ = 0;

 public:
  // join bookkeeping shared with core
  bool local_joined(int32_t set_id) const {
    auto it = process_sets_.find(set_id);
    return it != process_sets_.end() && it->second.joined.count(rank_) > 0;
  }
};

}  // namespace hvd
