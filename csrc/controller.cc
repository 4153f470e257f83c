#include "controller.h"

#include <algorithm>
#include <cstdio>
#include <cstring>
#include <sstream>

#include "core.h"
#include "logging.h"
#include "timeline.h"

namespace hvd {

namespace {

// Per-tensor shape block inside Response::tensor_shapes: [ndim, d0, d1, ...]
void AppendShape(std::vector<int64_t>& out, const std::vector<int64_t>& shape) {
  out.push_back((int64_t)shape.size());
  for (auto d : shape) out.push_back(d);
}

int64_t NumelOf(const std::vector<int64_t>& shape) {
  int64_t n = 1;
  for (auto d : shape) n *= d;
  return n;
}

std::string SetKey(int32_t set_id, const std::string& name) {
  return std::to_string(set_id) + ":" + name;
}

bool SameSignature(const Request& a, const Request& b) {
  return a.type == b.type && a.dtype == b.dtype && a.shape == b.shape &&
         a.root_rank == b.root_rank && a.reduce_op == b.reduce_op &&
         a.process_set_id == b.process_set_id && a.device == b.device &&
         a.group_key == b.group_key && a.group_size == b.group_size &&
         a.splits == b.splits;
}

// merge single-tensor responses of one group into one fused response
// (bypasses the fusion threshold — reference GroupTable semantics)
Response MergeGroup(std::vector<Response>& singles) {
  Response out = singles.front();
  for (size_t i = 1; i < singles.size(); ++i) {
    out.names.push_back(singles[i].names[0]);
    out.tensor_shapes.insert(out.tensor_shapes.end(),
                             singles[i].tensor_shapes.begin(),
                             singles[i].tensor_shapes.end());
  }
  return out;
}

bool IsFusableType(ResponseType t) {
  return t == ResponseType::ALLREDUCE || t == ResponseType::ADASUM;
}

}  // namespace

// ---------------------------------------------------------------------------
// ResponseCache
// ---------------------------------------------------------------------------

ResponseCache::State ResponseCache::Lookup(const Request& req) const {
  auto it = name_to_slot_.find(SetKey(req.process_set_id, req.name));
  if (it == name_to_slot_.end()) return State::MISS;
  const Entry& e = entries_[it->second];
  if (!e.live) return State::MISS;
  if (!SameSignature(e.request, req)) return State::INVALID;
  return State::HIT;
}

int ResponseCache::SlotOf(const std::string& key) const {
  auto it = name_to_slot_.find(key);
  if (it == name_to_slot_.end()) return -1;
  return entries_[it->second].live ? it->second : -1;
}

const Response& ResponseCache::Get(int slot) const { return entries_[slot].response; }
const Request& ResponseCache::GetRequest(int slot) const { return entries_[slot].request; }

int ResponseCache::AllocSlot() {
  if (!free_slots_.empty()) {
    int s = free_slots_.back();
    free_slots_.pop_back();
    return s;
  }
  if (entries_.size() < capacity_) {
    entries_.emplace_back();
    return (int)entries_.size() - 1;
  }
  // LRU eviction — deterministic across ranks because last_use ticks advance
  // identically (updates happen only on broadcast-ordered responses).
  int lru = -1;
  uint64_t best = UINT64_MAX;
  for (int i = 0; i < (int)entries_.size(); ++i) {
    if (entries_[i].live && entries_[i].last_use < best) {
      best = entries_[i].last_use;
      lru = i;
    }
  }
  if (lru >= 0) {
    Evict(lru);
    // Evict() queues the slot on the free list for its other callers; we
    // are handing it out RIGHT NOW — leaving it queued let the next
    // allocation reuse the same slot under capacity pressure, aliasing two
    // names onto one entry (fast-path deadlock: a rank votes a slot whose
    // cached response belongs to a different tensor)
    free_slots_.pop_back();
  }
  return lru >= 0 ? lru : 0;
}

void ResponseCache::EvictSet(int32_t process_set_id) {
  for (int i = 0; i < (int)entries_.size(); ++i)
    if (entries_[i].live && entries_[i].request.process_set_id == process_set_id)
      Evict(i);
}

void ResponseCache::Clear() {
  for (int i = 0; i < (int)entries_.size(); ++i) Evict(i);
}

void ResponseCache::Evict(int slot) {
  Entry& e = entries_[slot];
  if (!e.live) return;
  name_to_slot_.erase(SetKey(e.request.process_set_id, e.request.name));
  e.live = false;
  free_slots_.push_back(slot);
}

void ResponseCache::Put(const Response& response, const std::vector<Request>& reqs,
                        int32_t my_local_index) {
  if (response.type == ResponseType::JOIN || response.type == ResponseType::BARRIER ||
      response.type == ResponseType::ERROR || response.type == ResponseType::TUNE)
    return;
  // ALLTOALL caches only with a resolvable set-local index (the signature
  // needs this rank's send-split row out of the full matrix)
  int n_set = 0;
  if (response.type == ResponseType::ALLTOALL) {
    if (my_local_index < 0 || response.names.size() != 1) return;
    // tensor_sizes is the n x n split matrix
    while ((size_t)(n_set * n_set) < response.tensor_sizes.size()) ++n_set;
    if ((size_t)(n_set * n_set) != response.tensor_sizes.size()) return;
  }
  // Split a fused response into single-tensor cache entries.
  size_t nsizes_per = response.names.size()
                          ? response.tensor_sizes.size() / response.names.size()
                          : 0;
  const int64_t* shp = response.tensor_shapes.data();
  for (size_t i = 0; i < response.names.size(); ++i) {
    Response single;
    single.type = response.type;
    single.names = {response.names[i]};
    single.dtype = response.dtype;
    single.reduce_op = response.reduce_op;
    single.process_set_id = response.process_set_id;
    single.device = response.device;
    single.root_rank = response.root_rank;
    single.group_key = response.group_key;
    single.group_size = response.group_size;
    if (nsizes_per)
      single.tensor_sizes.assign(response.tensor_sizes.begin() + i * nsizes_per,
                                 response.tensor_sizes.begin() + (i + 1) * nsizes_per);
    std::vector<int64_t> shape;
    if (shp < response.tensor_shapes.data() + response.tensor_shapes.size()) {
      int64_t nd = *shp++;
      shape.assign(shp, shp + nd);
      shp += nd;
    }
    AppendShape(single.tensor_shapes, shape);

    Request sig;
    sig.type = (RequestType)response.type;
    sig.name = response.names[i];
    sig.dtype = response.dtype;
    sig.shape = (i < reqs.size() && !reqs[i].shape.empty()) ? reqs[i].shape : shape;
    sig.root_rank = response.root_rank;
    // The response carries the negotiated reduce_op (ConstructResponse copies
    // it from the first request); without it Min/Max/Product signatures would
    // Lookup as INVALID forever, defeating the cache fast path for those ops.
    sig.reduce_op = response.type == ResponseType::ADASUM ? ReduceOp::ADASUM
                                                          : response.reduce_op;
    if (i < reqs.size()) sig.reduce_op = reqs[i].reduce_op;
    sig.process_set_id = response.process_set_id;
    sig.device = response.device;
    sig.group_key = response.group_key;
    sig.group_size = response.group_size;
    if (response.type == ResponseType::ALLTOALL) {
      sig.type = RequestType::ALLTOALL;
      sig.splits.assign(
          response.tensor_sizes.begin() + (size_t)my_local_index * n_set,
          response.tensor_sizes.begin() + (size_t)(my_local_index + 1) * n_set);
    }

    std::string key = SetKey(sig.process_set_id, sig.name);
    auto it = name_to_slot_.find(key);
    int slot;
    if (it != name_to_slot_.end()) {
      slot = it->second;
    } else {
      slot = AllocSlot();
      name_to_slot_[key] = slot;
    }
    Entry& e = entries_[slot];
    e.request = sig;
    e.response = std::move(single);
    e.live = true;
    e.last_use = ++use_tick_;
  }
}

// ---------------------------------------------------------------------------
// Controller
// ---------------------------------------------------------------------------

Controller::Controller(StarComm* comm, int rank, int size, ControllerConfig cfg)
    : comm_(comm), rank_(rank), size_(size), cfg_(cfg) {
  cache_.set_capacity(cfg_.cache_capacity);
  check_stream_ = std::getenv("HOROVOD_CHECK_RESPONSE_STREAM") != nullptr;
  if (const char* tp = std::getenv("HOROVOD_TRACE_CYCLES")) {
    std::string path = std::string(tp) + "." + std::to_string(rank_);
    trace_ = std::fopen(path.c_str(), "w");
  }
  start_time_ = std::chrono::steady_clock::now();
  if (rank_ == 0 && size_ > 1 && std::getenv("HOROVOD_AUTOTUNE")) {
    const char* lp = std::getenv("HOROVOD_AUTOTUNE_LOG");
    const char* ot = std::getenv("HOROVOD_ONESHOT_THRESHOLD");
    autotuner_.reset(new Autotuner(cfg_.fusion_threshold_bytes,
                                   cfg_.cycle_time_ms,
                                   ot ? atoll(ot) : (int64_t)(4 << 20),
                                   lp ? lp : ""));
  }
  ProcessSetInfo global;
  global.id = 0;
  for (int r = 0; r < size; ++r) global.ranks.push_back(r);
  process_sets_[0] = std::move(global);
  last_stall_check_ = std::chrono::steady_clock::now();
}

int32_t Controller::AddProcessSet(const std::vector<int32_t>& ranks) {
  ProcessSetInfo info;
  info.id = next_set_id_++;
  info.ranks = ranks;
  std::sort(info.ranks.begin(), info.ranks.end());
  process_sets_[info.id] = std::move(info);
  return next_set_id_ - 1;
}

void Controller::RemoveProcessSet(int32_t id) {
  // evict the set's cache slots: stale live slots for an unknown set would
  // win the fast-path vote forever (every rank votes non-member = ready)
  cache_.EvictSet(id);
  process_sets_.erase(id);
}

ResponseList Controller::RunCycle(std::vector<Request> new_requests,
                                  bool shutdown_requested) {
  // -- 1. classify ----------------------------------------------------------
  auto tl = GetTimeline(State());
  std::vector<Request> slow;  // to coordinator this cycle
  std::vector<int> my_invalid_slots;
  for (auto& req : new_requests) {
    if (tl) tl->NegotiateStart(req.name);
    if (req.type == RequestType::JOIN || req.type == RequestType::BARRIER) {
      // never cached: join/barrier are stateful.  (alltoall caches too —
      // the send-split row is part of the signature, so changed splits
      // invalidate and renegotiate.)
      if (req.type == RequestType::JOIN)
        my_joined_.insert(req.process_set_id);
      slow.push_back(std::move(req));
      continue;
    }
    auto state = cache_enabled_ ? cache_.Lookup(req)
                                : ResponseCache::State::MISS;
    if (state == ResponseCache::State::HIT) {
      cached_pending_.push_back(std::move(req));
    } else {
      if (state == ResponseCache::State::INVALID) {
        int slot = cache_.SlotOf(SetKey(req.process_set_id, req.name));
        if (slot >= 0) my_invalid_slots.push_back(slot);
      }
      slow.push_back(std::move(req));
    }
  }

  // re-classify pending hits whose cache slot was since evicted/replaced
  // (capacity-pressure Put evictions would otherwise strand them forever).
  // A cache-off TUNE also drains ALL pending to the slow path: a request
  // classified HIT just before the flip would otherwise keep fast-path
  // voting while its peers renegotiate — cross-rank deadlock.
  for (auto it = cached_pending_.begin(); it != cached_pending_.end();) {
    if (!cache_enabled_ ||
        cache_.Lookup(*it) != ResponseCache::State::HIT) {
      slow.push_back(*it);
      it = cached_pending_.erase(it);
    } else {
      ++it;
    }
  }

  // -- 2. bitvector rounds --------------------------------------------------
  const size_t nbits = cfg_.cache_capacity;
  const size_t nbytes = (nbits + 7) / 8;
  ResponseList result;

  if (size_ > 1) {
    const size_t tail = check_stream_ ? sizeof(uint64_t) : 0;
    std::string vecA(nbytes + 1 + tail, '\0');
    // A rank votes "ready" for a slot if it has the tensor queued OR it is
    // not a member of the slot's process set (so subset tensors can take the
    // fast path under a global AND).
    std::vector<bool> mine(cache_.num_slots(), false);
    for (auto& req : cached_pending_) {
      int slot = cache_.SlotOf(SetKey(req.process_set_id, req.name));
      if (slot >= 0) mine[slot] = true;
    }
    for (size_t s = 0; s < cache_.num_slots(); ++s) {
      bool vote = false;
      if (cache_.slot_live((int)s)) {
        const Request& sig = cache_.GetRequest((int)s);
        auto it = process_sets_.find(sig.process_set_id);
        bool member = it != process_sets_.end() && it->second.contains(rank_);
        // a joined rank votes ready for every slot of that set — it will
        // zero-substitute, so steady-state CACHED tensors of busier peers
        // keep firing (reference: controller.cc:130-134 records a hit for
        // all bits when joined; without this, join + cache fast path
        // deadlocks)
        vote = mine[s] || !member || my_joined_.count(sig.process_set_id);
      }
      if (vote) vecA[s / 8] |= (char)(1 << (s % 8));
    }
    if (shutdown_requested) vecA[nbytes] |= 1;
    if (tail)
      std::memcpy(&vecA[nbytes + 1], &response_hash_, sizeof(uint64_t));
    comm_->BitAnd(vecA, tail);

    std::string vecB(nbytes + 1, '\0');
    for (int slot : my_invalid_slots) vecB[slot / 8] |= (char)(1 << (slot % 8));
    bool need_slow = !slow.empty() || !inflight_.empty() || pending_tune_;
    if (need_slow) vecB[nbytes] |= 1;
    comm_->BitOr(vecB);

    result.shutdown = (vecA[nbytes] & 1) != 0;
    bool any_slow = (vecB[nbytes] & 1) != 0;

    // -- 3. evictions + fast path -------------------------------------------
    for (size_t s = 0; s < cache_.num_slots(); ++s) {
      if (vecB[s / 8] & (1 << (s % 8))) {
        cache_.Evict((int)s);
        // pending requests that were hits against the evicted slot must go to
        // the coordinator now.
        for (auto it = cached_pending_.begin(); it != cached_pending_.end();) {
          if (cache_.Lookup(*it) != ResponseCache::State::HIT) {
            slow.push_back(*it);
            it = cached_pending_.erase(it);
            if (!any_slow) { any_slow = true; }  // we now need the slow round
          } else {
            ++it;
          }
        }
      }
    }
    // NOTE: if an eviction forced `slow` entries after the OR round already
    // ran, the OR flag may be false on every rank.  Both evictor and owner
    // see the same OR bits, so every rank whose pending got displaced flags
    // the next cycle via non-empty `slow`/`inflight_` — we just hold them.
    // collect common-hit slots; grouped tensors only fire when the whole
    // group is commonly ready (strict GroupTable semantics)
    std::vector<int> hit_slots;
    std::unordered_map<std::string, std::vector<int>> group_slots;
    for (size_t s = 0; s < cache_.num_slots(); ++s) {
      if (!(vecA[s / 8] & (1 << (s % 8)))) continue;
      if (vecB[s / 8] & (1 << (s % 8))) continue;  // invalidated this cycle
      if (!cache_.slot_live((int)s)) continue;
      const Response& r = cache_.Get((int)s);
      if (!r.group_key.empty() && r.group_size > 1)
        group_slots[SetKey(r.process_set_id, r.group_key)].push_back((int)s);
      else
        hit_slots.push_back((int)s);
    }
    std::deque<Response> fast;
    auto consume = [&](int s) {
      const Response& r = cache_.Get(s);
      for (auto it = cached_pending_.begin(); it != cached_pending_.end(); ++it) {
        if (SetKey(it->process_set_id, it->name) ==
            SetKey(r.process_set_id, r.names[0])) {
          cached_pending_.erase(it);
          break;
        }
      }
      return r;
    };
    for (int s : hit_slots) fast.push_back(consume(s));
    // groups in deterministic order (min slot): unordered_map iteration
    // order differs across ranks, and the response stream must be
    // bit-identical everywhere (it serializes the RCCL call order)
    std::vector<std::vector<int>*> ordered_groups;
    for (auto& kv : group_slots) {
      std::sort(kv.second.begin(), kv.second.end());
      ordered_groups.push_back(&kv.second);
    }
    std::sort(ordered_groups.begin(), ordered_groups.end(),
              [](auto* a, auto* b) { return (*a)[0] < (*b)[0]; });
    for (auto* slots_p : ordered_groups) {
      auto& slots = *slots_p;
      if ((int32_t)slots.size() != cache_.Get(slots[0]).group_size)
        continue;  // incomplete group: stays pending, re-votes next cycle
      std::vector<Response> singles;
      for (int s : slots) singles.push_back(consume(s));
      fast.push_back(MergeGroup(singles));
    }
    auto fast_fused = FuseResponses(fast);
    for (auto& r : fast_fused) result.responses.push_back(std::move(r));

    // -- 4. slow path ---------------------------------------------------------
    if (any_slow || (vecB[nbytes] & 1)) {
      RequestList rl;
      rl.requests = slow;
      std::string payload;
      rl.Serialize(payload);
      auto gathered = comm_->Gather(payload);
      for (auto& req : slow) inflight_.insert(SetKey(req.process_set_id, req.name));

      std::string resp_payload;
      if (rank_ == 0) {
        auto responses = CoordinatorProcess(gathered);
        ResponseList out;
        out.responses = std::move(responses);
        out.Serialize(resp_payload);
        resp_payload = comm_->Bcast(resp_payload);
      } else {
        resp_payload = comm_->Bcast("");
      }
      auto slow_result = ResponseList::Deserialize(
          resp_payload.data(), resp_payload.data() + resp_payload.size());
      for (auto& resp : slow_result.responses) {
        for (auto& n : resp.names) inflight_.erase(SetKey(resp.process_set_id, n));
        if (resp.type == ResponseType::JOIN || resp.type == ResponseType::BARRIER)
          inflight_.erase(SetKey(resp.process_set_id,
                                 resp.type == ResponseType::JOIN ? "join" : "barrier"));
        if (resp.type == ResponseType::JOIN)
          my_joined_.erase(resp.process_set_id);  // everyone joined: reset
        if (cache_enabled_) {
          int32_t li = -1;
          auto its = process_sets_.find(resp.process_set_id);
          if (its != process_sets_.end()) li = its->second.local_index(rank_);
          cache_.Put(resp, {}, li);
        }
        result.responses.push_back(std::move(resp));
      }
    }
    // (A non-empty `slow` always implies any_slow: the owning rank set the
    // OR flag in the same cycle, so the branch above always runs.)
  } else {
    // Single process: answer everything immediately, no sockets.
    result.shutdown = shutdown_requested;
    std::deque<Response> ready;
    for (auto& req : slow) {
      auto key = SetKey(req.process_set_id, req.name);
      PendingTensor pt;
      pt.requests = {req};
      Response r = ConstructResponse(key, pt);
      if (r.type == ResponseType::JOIN) r.last_joined_rank = rank_;
      ready.push_back(std::move(r));
    }
    for (auto& req : cached_pending_) {
      PendingTensor pt;
      pt.requests = {req};
      ready.push_back(ConstructResponse(SetKey(req.process_set_id, req.name), pt));
    }
    cached_pending_.clear();
    auto fused = FuseResponses(ready);
    for (auto& r : fused) result.responses.push_back(std::move(r));
  }

  if (trace_) {
    std::fprintf(trace_, "c%llu new=%zu slow_held=%zu pend=[",
                 (unsigned long long)trace_cycle_++, new_requests.size(),
                 inflight_.size());
    for (auto& pr : cached_pending_)
      std::fprintf(trace_, "%s/%d,", pr.name.c_str(),
                   cache_.SlotOf(SetKey(pr.process_set_id, pr.name)));
    std::fprintf(trace_, "] resp=");
    for (auto& r : result.responses) {
      std::fprintf(trace_, "%d:", (int)r.type);
      for (auto& n : r.names) std::fprintf(trace_, "%s,", n.c_str());
      std::fprintf(trace_, ";");
    }
    std::fprintf(trace_, " cache=%d fus=%lld cyc=%.3f\n", (int)cache_enabled_,
                 (long long)cfg_.fusion_threshold_bytes, cfg_.cycle_time_ms);
    std::fflush(trace_);
  }
  if (check_stream_) {
    // FNV-1a over each response's wire serialization, in execution order
    for (auto& resp : result.responses) {
      std::string ser;
      resp.Serialize(ser);
      for (unsigned char c : ser) {
        response_hash_ ^= c;
        response_hash_ *= 1099511628211ull;
      }
    }
  }

  // ---- autotuner (rank 0): score this cycle's reduced bytes; a ready
  // proposal is published next cycle as a TUNE response (slow path forced
  // via pending_tune_).
  if (autotuner_) {
    int64_t bytes = 0;
    for (auto& resp : result.responses) {
      if (resp.type != ResponseType::ALLREDUCE &&
          resp.type != ResponseType::ADASUM)
        continue;
      const int64_t* p = resp.tensor_shapes.data();
      const int64_t* end = p + resp.tensor_shapes.size();
      while (p < end) {
        int64_t nd = *p++;
        int64_t n = 1;
        for (int64_t i = 0; i < nd; ++i) n *= *p++;
        bytes += n * (int64_t)DataTypeSize(resp.dtype);
      }
    }
    double now = std::chrono::duration<double>(std::chrono::steady_clock::now() -
                                               start_time_)
                     .count();
    bool propose = autotuner_->done() ? autotuner_->Watch(bytes, now)
                                      : autotuner_->Record(bytes, now);
    if (propose) pending_tune_ = true;
  }
  return result;
}

std::vector<Response> Controller::CoordinatorProcess(
    const std::vector<std::string>& gathered) {
  // Feed new requests into the pending table.
  auto& arrival_order = arrival_order_;
  for (int r = 0; r < (int)gathered.size(); ++r) {
    if (gathered[r].empty()) continue;
    auto rl = RequestList::Deserialize(gathered[r].data(),
                                       gathered[r].data() + gathered[r].size());
    for (auto& req : rl.requests) {
      auto& set = process_sets_.at(req.process_set_id);
      if (req.type == RequestType::JOIN) {
        set.joined.insert(req.rank);
        set.last_joined_rank = req.rank;
        continue;
      }
      std::string key = SetKey(req.process_set_id, req.name);
      auto it = table_.find(key);
      if (it == table_.end()) {
        PendingTensor pt;
        pt.have.assign(set.ranks.size(), false);
        pt.first_seen = std::chrono::steady_clock::now();
        it = table_.emplace(key, std::move(pt)).first;
        arrival_order.push_back(key);
      }
      int li = set.local_index(req.rank);
      if (li >= 0 && !it->second.have[li]) {
        it->second.have[li] = true;
        it->second.count++;
        it->second.requests.push_back(req);
      }
    }
  }

  // Find complete tensors in arrival order.
  for (auto ko = arrival_order.begin(); ko != arrival_order.end();) {
    auto it = table_.find(*ko);
    if (it == table_.end()) {
      ko = arrival_order.erase(ko);
      continue;
    }
    auto& pt = it->second;
    int32_t set_id = pt.requests.front().process_set_id;
    auto& set = process_sets_.at(set_id);
    int needed = (int)set.ranks.size();
    if (pt.requests.front().type != RequestType::BARRIER) {
      // joined ranks are counted as implicitly ready (zero contribution)
      int joined_needed = 0;
      for (auto jr : set.joined)
        if (set.contains(jr) && !pt.have[set.local_index(jr)]) joined_needed++;
      needed -= joined_needed;
    }
    if (pt.count >= needed) {
      Response r = ConstructResponse(it->first, pt);
      if (!r.group_key.empty() && r.group_size > 1) {
        std::string gk = SetKey(r.process_set_id, r.group_key);
        if (r.type == ResponseType::ERROR) {
          // poisoned group: release anything already held as individual
          // responses so peers error out instead of hanging on the merge
          poisoned_groups_.insert(gk);
          auto hit = group_hold_.find(gk);
          if (hit != group_hold_.end()) {
            for (auto& held : hit->second)
              ready_responses_.push_back(std::move(held));
            group_hold_.erase(hit);
          }
          ready_responses_.push_back(std::move(r));
        } else if (poisoned_groups_.count(gk)) {
          ready_responses_.push_back(std::move(r));
        } else {
          // strict group semantics (reference GroupTable): hold until every
          // member of the group is ready, then emit ONE fused response.
          auto& hold = group_hold_[gk];
          hold.push_back(std::move(r));
          if ((int32_t)hold.size() == hold.front().group_size) {
            ready_responses_.push_back(MergeGroup(hold));
            group_hold_.erase(gk);
            poisoned_groups_.erase(gk);
          }
        }
      } else {
        ready_responses_.push_back(std::move(r));
      }
      table_.erase(it);
      ko = arrival_order.erase(ko);
    } else {
      ++ko;
    }
  }

  // JOIN completion: all members of a set joined -> emit JOIN response and
  // reset the joined state (reference: controller join handling).
  for (auto& kv : process_sets_) {
    auto& set = kv.second;
    if (!set.ranks.empty() && set.joined.size() == set.ranks.size()) {
      Response jr;
      jr.type = ResponseType::JOIN;
      jr.process_set_id = set.id;
      jr.last_joined_rank = set.last_joined_rank;
      ready_responses_.push_back(std::move(jr));
      set.joined.clear();
      set.last_joined_rank = -1;
    }
  }

  CheckForStalledTensors();
  auto out = FuseResponses(ready_responses_);
  if (pending_tune_ && autotuner_) {
    Response t;
    t.type = ResponseType::TUNE;
    auto p = autotuner_->current();
    t.tensor_sizes = {p.fusion_bytes, (int64_t)(p.cycle_time_ms * 1000.0),
                      p.oneshot_threshold, p.cache_enabled ? 1 : 0};
    out.push_back(std::move(t));
    pending_tune_ = false;
  }
  return out;
}

Response Controller::ConstructResponse(const std::string& key, PendingTensor& pt) {
  auto& reqs = pt.requests;
  const Request& first = reqs.front();
  Response resp;
  resp.names = {first.name};
  resp.dtype = first.dtype;
  resp.reduce_op = first.reduce_op;
  resp.process_set_id = first.process_set_id;
  resp.device = first.device;
  resp.root_rank = first.root_rank;
  resp.group_key = first.group_key;
  resp.group_size = first.group_size;

  // Validate cross-rank consistency (reference: controller.cc
  // ConstructResponse:496-843 error text behavior).
  std::ostringstream err;
  for (size_t i = 1; i < reqs.size(); ++i) {
    const Request& r = reqs[i];
    if (r.type != first.type) {
      err << "Mismatched collective operations submitted for tensor " << first.name;
      break;
    }
    if (r.dtype != first.dtype) {
      err << "Mismatched data types for tensor " << first.name << ": one rank sent "
          << DataTypeName(first.dtype) << ", another sent " << DataTypeName(r.dtype);
      break;
    }
    if (r.root_rank != first.root_rank) {
      err << "Mismatched root ranks for broadcast of tensor " << first.name;
      break;
    }
    if (r.reduce_op != first.reduce_op) {
      err << "Mismatched reduction ops for tensor " << first.name;
      break;
    }
    bool shape_must_match = first.type == RequestType::ALLREDUCE ||
                            first.type == RequestType::ADASUM ||
                            first.type == RequestType::BROADCAST;
    if (shape_must_match && r.shape != first.shape) {
      err << "Mismatched tensor shapes for " << first.name;
      break;
    }
    if ((first.type == RequestType::ALLGATHER ||
         first.type == RequestType::REDUCESCATTER) &&
        r.shape.size() == first.shape.size() && !r.shape.empty()) {
      for (size_t d = 1; d < r.shape.size(); ++d) {
        if (r.shape[d] != first.shape[d]) {
          err << "Mismatched non-first dimensions for tensor " << first.name;
          break;
        }
      }
    }
  }
  if (!err.str().empty()) {
    resp.type = ResponseType::ERROR;
    resp.error_msg = err.str();
    return resp;
  }

  auto& set = process_sets_.at(first.process_set_id);
  switch (first.type) {
    case RequestType::ALLREDUCE:
      resp.type = first.reduce_op == ReduceOp::ADASUM ? ResponseType::ADASUM
                                                      : ResponseType::ALLREDUCE;
      AppendShape(resp.tensor_shapes, first.shape);
      break;
    case RequestType::ADASUM:
      resp.type = ResponseType::ADASUM;
      AppendShape(resp.tensor_shapes, first.shape);
      break;
    case RequestType::BROADCAST:
      resp.type = ResponseType::BROADCAST;
      AppendShape(resp.tensor_shapes, first.shape);
      break;
    case RequestType::BARRIER:
      resp.type = ResponseType::BARRIER;
      break;
    case RequestType::ALLGATHER:
    case RequestType::REDUCESCATTER: {
      resp.type = first.type == RequestType::ALLGATHER ? ResponseType::ALLGATHER
                                                       : ResponseType::REDUCESCATTER;
      AppendShape(resp.tensor_shapes, first.shape);
      // first-dimension contribution per set-local rank (0 for joined /
      // missing ranks)
      resp.tensor_sizes.assign(set.ranks.size(), 0);
      for (auto& r : reqs) {
        int li = set.local_index(r.rank);
        if (li >= 0) resp.tensor_sizes[li] = r.shape.empty() ? 0 : r.shape[0];
      }
      break;
    }
    case RequestType::ALLTOALL: {
      resp.type = ResponseType::ALLTOALL;
      AppendShape(resp.tensor_shapes, first.shape);
      // Full send-split matrix, row = set-local sender index (reference:
      // AlltoallOp::PrepareOutputAndParams exchanges splits via the
      // controller, collective_operations.h:200-260).
      int n = (int)set.ranks.size();
      resp.tensor_sizes.assign((size_t)n * n, 0);
      for (auto& r : reqs) {
        int li = set.local_index(r.rank);
        if (li < 0) continue;
        if ((int)r.splits.size() == n) {
          for (int j = 0; j < n; ++j) resp.tensor_sizes[li * n + j] = r.splits[j];
        } else {
          int64_t first_dim = r.shape.empty() ? 0 : r.shape[0];
          for (int j = 0; j < n; ++j) resp.tensor_sizes[li * n + j] = first_dim / n;
        }
      }
      break;
    }
    case RequestType::JOIN:
      resp.type = ResponseType::JOIN;
      break;
  }
  return resp;
}

std::vector<Response> Controller::FuseResponses(std::deque<Response>& queue) {
  // Greedy fusion with lookahead (reference: controller.cc FuseResponses
  // 901-1091): pull the front response, then scan the rest of the queue for
  // compatible tensors until the threshold fills.
  std::vector<Response> out;
  while (!queue.empty()) {
    Response r = std::move(queue.front());
    queue.pop_front();
    if (!IsFusableType(r.type) || r.names.size() != 1) {
      out.push_back(std::move(r));
      continue;
    }
    auto shape_of = [](const Response& resp) {
      std::vector<int64_t> s;
      const int64_t* p = resp.tensor_shapes.data();
      int64_t nd = resp.tensor_shapes.empty() ? 0 : *p++;
      s.assign(p, p + nd);
      return s;
    };
    int64_t bytes = AlignedElems(NumelOf(shape_of(r))) * (int64_t)DataTypeSize(r.dtype);
    for (auto it = queue.begin();
         it != queue.end() && bytes < cfg_.fusion_threshold_bytes;) {
      if (it->type == r.type && it->dtype == r.dtype && it->device == r.device &&
          it->process_set_id == r.process_set_id &&
          it->reduce_op == r.reduce_op && it->names.size() == 1) {
        int64_t add =
            AlignedElems(NumelOf(shape_of(*it))) * (int64_t)DataTypeSize(r.dtype);
        if (bytes + add > cfg_.fusion_threshold_bytes && bytes > 0) {
          ++it;
          continue;
        }
        bytes += add;
        r.names.push_back(it->names[0]);
        r.tensor_shapes.insert(r.tensor_shapes.end(), it->tensor_shapes.begin(),
                               it->tensor_shapes.end());
        it = queue.erase(it);
      } else {
        ++it;
      }
    }
    out.push_back(std::move(r));
  }
  return out;
}

void Controller::CheckForStalledTensors() {
  auto now = std::chrono::steady_clock::now();
  if (std::chrono::duration<double>(now - last_stall_check_).count() <
      cfg_.stall_warning_sec)
    return;
  last_stall_check_ = now;
  for (auto& kv : table_) {
    auto& pt = kv.second;
    double age = std::chrono::duration<double>(now - pt.first_seen).count();
    if (age < cfg_.stall_warning_sec) continue;
    auto& set = process_sets_.at(pt.requests.front().process_set_id);
    std::ostringstream missing;
    for (size_t i = 0; i < pt.have.size(); ++i)
      if (!pt.have[i]) missing << set.ranks[i] << " ";
    HVD_LOG(WARNING, "tensor %s stalled for %.0fs; waiting on ranks: %s",
            kv.first.c_str(), age, missing.str().c_str());
    if (cfg_.stall_shutdown_sec > 0 && age > cfg_.stall_shutdown_sec) {
      // reference: HOROVOD_STALL_SHUTDOWN_TIME_SECONDS aborts the job when a
      // stall persists (stall_inspector.h:30-97)
      throw std::runtime_error("stalled tensor " + kv.first + " exceeded "
                               "HOROVOD_STALL_SHUTDOWN_TIME_SECONDS; aborting");
    }
  }
}

}  // namespace hvd
