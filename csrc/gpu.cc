#include "gpu.h"

#include <hip/hip_runtime_api.h>
#include <rccl/rccl.h>
#include <roctracer/roctx.h>

#include <c10/hip/HIPCachingAllocator.h>
#include <c10/hip/HIPGuard.h>
#include <c10/hip/HIPStream.h>

#include <algorithm>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <mutex>
#include <stdexcept>
#include <thread>
#include <unordered_map>

#include "core.h"
#include "kernels.h"
#include "logging.h"
#include "timeline.h"

namespace hvd {
namespace gpu {

namespace {

#define HIP_CHECK(cmd)                                                      \
  do {                                                                      \
    hipError_t e_ = (cmd);                                                  \
    if (e_ != hipSuccess)                                                   \
      throw std::runtime_error(std::string("HIP error: ") +                 \
                               hipGetErrorString(e_) + " at " #cmd);        \
  } while (0)

#define RCCL_CHECK(cmd)                                                     \
  do {                                                                      \
    ncclResult_t r_ = (cmd);                                                \
    if (r_ != ncclSuccess)                                                  \
      throw std::runtime_error(std::string("RCCL error: ") +                \
                               ncclGetErrorString(r_) + " at " #cmd);       \
  } while (0)

ncclDataType_t ToNccl(DataType t) {
  switch (t) {
    case DataType::HVD_UINT8: return ncclUint8;
    case DataType::HVD_INT8: return ncclInt8;
    case DataType::HVD_INT32: return ncclInt32;
    case DataType::HVD_INT64: return ncclInt64;
    case DataType::HVD_FLOAT16: return ncclFloat16;
    case DataType::HVD_FLOAT32: return ncclFloat32;
    case DataType::HVD_FLOAT64: return ncclFloat64;
    case DataType::HVD_BFLOAT16: return ncclBfloat16;
    case DataType::HVD_BOOL: return ncclUint8;
    default:
      throw std::runtime_error("horovod_amd: dtype unsupported by RCCL");
  }
}

ncclRedOp_t ToNcclOp(ReduceOp op) {
  switch (op) {
    case ReduceOp::SUM:
    case ReduceOp::AVERAGE: return ncclSum;  // average via postscale
    case ReduceOp::MIN: return ncclMin;
    case ReduceOp::MAX: return ncclMax;
    case ReduceOp::PRODUCT: return ncclProd;
    default: throw std::runtime_error("horovod_amd: bad reduce op for RCCL");
  }
}

// One-shot allreduce shared-memory window for one (device, process set):
// my coarse staging (slots x threshold) + fine-grained flag page, plus the
// IPC-opened peer windows, set-local indexed.  See oneshot.hip for the
// device protocol.
struct OneshotCtx {
  bool ready = false;
  bool failed = false;
  char* my_staging = nullptr;
  unsigned long long* my_flags = nullptr;
  void* peer_staging[kOneshotMaxRanks] = {};
  void* peer_flags[kOneshotMaxRanks] = {};
  std::vector<void*> opened;  // to hipIpcCloseMemHandle on shutdown
  unsigned long long seq = 0;
  int64_t slot_bytes = 0;
  int nslots = 2;
  int n = 0, li = -1;
};

struct DeviceCtx {
  int device = -1;
  // Dedicated comm stream from torch's pool.  NORMAL priority by default:
  // on MI355X a high-priority HSA queue that is stalled on a cross-stream
  // barrier (waiting for gradient producers) degrades the compute queue's
  // throughput ~2.3x (measured: 27-30 ms/step vs 14.5 at normal priority,
  // ResNet-50 hooked bench).  HOROVOD_HIGH_PRIORITY_COMM_STREAM=1 opts back
  // in for latency-critical small-model workloads.
  c10::hip::HIPStream stream;
  at::Tensor fusion_buffer;    // persistent, byte-typed
  at::Tensor adasum_buffer;    // [set_size x fused] gather space for adasum
  at::Tensor dots_buffer;      // adasum per-tensor {dot,|a|2,|b|2} doubles
  std::unordered_map<int32_t, ncclComm_t> comms;  // process_set -> comm
  std::unordered_map<int32_t, OneshotCtx> oneshot;  // process_set -> window
  // hierarchical/torus allreduce sub-communicators (global set only)
  struct HierCtx {
    ncclComm_t local = nullptr;  // ranks on my node
    ncclComm_t cross = nullptr;  // same local_rank across nodes
    bool ready = false;
  };
  std::unordered_map<int32_t, HierCtx> hier;

  explicit DeviceCtx(int dev)
      : device(dev),
        stream(c10::hip::getStreamFromPool(
            std::getenv("HOROVOD_HIGH_PRIORITY_COMM_STREAM") != nullptr,
            dev)) {}
};

struct PendingOp {
  hipEvent_t done_event = nullptr;
  std::vector<TensorTableEntry> entries;
  std::vector<hipEvent_t> ready_events;
  int device = -1;
  int64_t start_us = 0;
  std::string activity;
  ncclComm_t comm = nullptr;  // for the async-error watchdog
};

struct Finalizer {
  std::mutex mu;
  std::condition_variable cv;
  std::deque<PendingOp> queue;
  std::thread thread;
  bool stop = false;
  bool started = false;
} g_finalizer;

std::mutex g_ctx_mu;
std::unordered_map<int, DeviceCtx*> g_ctx;
std::atomic<bool> g_rccl_used{false};
// Set once any communicator is aborted (async RCCL error or TCP peer loss).
// Every subsequent GPU op fails fast with ABORTED — the signal elastic
// recovery converts into HorovodInternalError (reference:
// nccl_operations.cc:56-147 commDestroyOrAbort + elastic_restart_).
std::atomic<bool> g_comm_failed{false};
// guards every DeviceCtx::comms map (bg thread inserts, finalizer aborts)
std::mutex g_comms_mu;

bool CommHasAsyncError(ncclComm_t comm) {
  if (!comm) return false;
  ncclResult_t async_err = ncclSuccess;
  if (ncclCommGetAsyncError(comm, &async_err) != ncclSuccess) return true;
  return async_err != ncclSuccess && async_err != ncclInProgress;
}

// hipEvent pool (reference: GPUContext event pool with prepopulation,
// hip_operations.cc:5-80) — create/destroy cost ~2 us each adds up at 161
// events/step.
std::mutex g_event_mu;
std::vector<hipEvent_t> g_event_pool;

hipEvent_t AcquireEvent() {
  {
    std::lock_guard<std::mutex> g(g_event_mu);
    if (!g_event_pool.empty()) {
      hipEvent_t ev = g_event_pool.back();
      g_event_pool.pop_back();
      return ev;
    }
  }
  hipEvent_t ev;
  HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
  return ev;
}

void ReleaseEvent(hipEvent_t ev) {
  std::lock_guard<std::mutex> g(g_event_mu);
  if (g_event_pool.size() < 1024)
    g_event_pool.push_back(ev);
  else
    (void)hipEventDestroy(ev);
}

void AbortAllCommsLocked(const char* why);

// Poll-based stream sync with a deadline: the RS-VHDD Adasum path blocks
// the background thread mid-Execute, where the finalizer watchdog cannot
// see a hung collective — a bounded wait turns a wedged exchange into a
// failed op instead of a dead process.
void SyncStreamBounded(hipStream_t stream, double timeout_sec = 300.0) {
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::duration<double>(timeout_sec);
  hipError_t e;
  while ((e = hipStreamQuery(stream)) == hipErrorNotReady) {
    if (g_comm_failed)
      throw std::runtime_error(
          "HorovodInternalError: comm aborted during Adasum exchange");
    if (std::chrono::steady_clock::now() > deadline)
      throw std::runtime_error("Adasum VHDD exchange timed out");
    std::this_thread::sleep_for(std::chrono::microseconds(200));
  }
  if (e != hipSuccess)
    throw std::runtime_error(std::string("hipStreamQuery: ") +
                             hipGetErrorString(e));
}

// Standing async-error watchdog: the finalizer only polls while an op is
// pending, but a peer can die while the bg thread is blocked inside a
// synchronous exchange (RS-VHDD) with nothing queued.  This thread checks
// every live comm's async state at 100 ms cadence.
struct CommWatchdog {
  std::thread thread;
  std::atomic<bool> stop{false};
  bool started = false;
} g_watchdog;

void WatchdogLoop() {
  while (!g_watchdog.stop) {
    std::this_thread::sleep_for(std::chrono::milliseconds(100));
    if (g_comm_failed) continue;
    std::lock_guard<std::mutex> gc(g_ctx_mu);
    std::lock_guard<std::mutex> g(g_comms_mu);
    bool bad = false;
    for (auto& kv : g_ctx) {
      for (auto& ck : kv.second->comms)
        if (CommHasAsyncError(ck.second)) bad = true;
      if (bad) break;
    }
    if (bad) AbortAllCommsLocked("RCCL async error (standing watchdog)");
  }
}

void EnsureWatchdog() {
  if (!g_watchdog.started) {
    g_watchdog.stop = false;
    g_watchdog.thread = std::thread(WatchdogLoop);
    g_watchdog.started = true;
  }
}

void FinalizerLoop() {
  while (true) {
    PendingOp op;
    {
      std::unique_lock<std::mutex> lk(g_finalizer.mu);
      g_finalizer.cv.wait(lk,
                          [] { return g_finalizer.stop || !g_finalizer.queue.empty(); });
      if (g_finalizer.queue.empty()) {
        if (g_finalizer.stop) return;
        continue;
      }
      op = std::move(g_finalizer.queue.front());
      g_finalizer.queue.pop_front();
    }
    // sleep-poll completion: a hot hipEventQuery/hipEventSynchronize spin
    // contends with the training thread's kernel launches on HIP runtime
    // locks (measured: +150% step time under 161-tensor gradient flow);
    // 50 us sleeps make the query rate negligible while keeping completion
    // latency far below a bucket's comm time.
    //
    // Watchdog (reference: nccl_operations.cc AsyncErrorCheck): every ~10 ms
    // of waiting, poll ncclCommGetAsyncError on this op's comm and the
    // global failure flag; on error, ncclCommAbort every comm so the hung
    // collective unblocks, then fail the entries with ABORTED so user-side
    // synchronize() raises HorovodInternalError instead of hanging.
    hipError_t e;
    int spins = 0;
    bool comm_dead = false;
    auto fail_deadline = std::chrono::steady_clock::time_point::max();
    while ((e = hipEventQuery(op.done_event)) == hipErrorNotReady) {
      std::this_thread::sleep_for(std::chrono::microseconds(50));
      if (++spins % 200 != 0) continue;  // async checks every ~10 ms
      auto now = std::chrono::steady_clock::now();
      if (comm_dead) {
        if (now > fail_deadline) break;  // abort didn't unblock the stream
        continue;
      }
      if (g_comm_failed || CommHasAsyncError(op.comm)) {
        {
          std::lock_guard<std::mutex> gc(g_ctx_mu);
          std::lock_guard<std::mutex> g(g_comms_mu);
          AbortAllCommsLocked("RCCL async error detected by finalizer");
        }
        comm_dead = true;
        // give ncclCommAbort 2 s to terminate the enqueued kernel cleanly
        fail_deadline = now + std::chrono::seconds(2);
      }
    }
    Status s;
    if (comm_dead) {
      s = Status::Aborted(
          "RCCL communicator aborted (peer failure or async error)");
    } else if (e == hipSuccess) {
      s = Status::OK();
    } else {
      s = Status::UnknownError(std::string("hipEventSynchronize: ") +
                               hipGetErrorString(e));
    }
    for (auto& entry : op.entries)
      if (entry.callback) entry.callback(s, entry);
    if (e != hipErrorNotReady) {
      // only recycle events that actually signaled; a wedged event must not
      // re-enter the pool (it would poison a future op)
      ReleaseEvent(op.done_event);
      for (auto ev : op.ready_events) ReleaseEvent(ev);
    }
    auto& st = State();
    auto tl = GetTimeline(st);
    if (tl && !op.entries.empty())
      tl->Activity(op.entries[0].name, op.activity, op.start_us, tl->NowUs());
  }
}

void EnsureFinalizer() {
  std::lock_guard<std::mutex> g(g_finalizer.mu);
  if (!g_finalizer.started) {
    g_finalizer.thread = std::thread(FinalizerLoop);
    g_finalizer.started = true;
  }
}

DeviceCtx& GetCtx(int device) {
  std::lock_guard<std::mutex> g(g_ctx_mu);
  auto it = g_ctx.find(device);
  if (it != g_ctx.end()) return *it->second;
  c10::hip::HIPGuard guard(device);
  auto* ctx = new DeviceCtx(device);
  g_ctx[device] = ctx;
  return *ctx;
}

// uniqueId exchange over the TCP star — every global rank participates so
// the lock-step framing holds even when rank 0 is not a set member.
std::string ExchangeUniqueId(GlobalState& st, int leader,
                             const std::string& payload) {
  auto gathered = st.comm.Gather(payload);
  std::string idb;
  if (st.comm.is_root() && leader < (int)gathered.size()) idb = gathered[leader];
  return st.comm.Bcast(idb);
}

// Abort every live communicator (called with g_ctx_mu AND g_comms_mu held,
// in that order).  After this, all GPU collectives fail fast until the
// next elastic re-init.
void AbortAllCommsLocked(const char* why) {
  if (g_comm_failed.exchange(true)) return;  // once
  HVD_LOG(ERROR, "aborting all RCCL comms: %s", why);
  for (auto& kv : g_ctx) {
    for (auto& ck : kv.second->comms) (void)ncclCommAbort(ck.second);
    kv.second->comms.clear();
    for (auto& hk : kv.second->hier) {
      if (hk.second.local) (void)ncclCommAbort(hk.second.local);
      if (hk.second.cross) (void)ncclCommAbort(hk.second.cross);
    }
    kv.second->hier.clear();
    // raise the one-shot abort word (flags[3]) so spin-wait kernels exit
    // instead of wedging the GPU on a dead peer (SDMA write proceeds even
    // while compute is spinning)
    for (auto& ok : kv.second->oneshot) {
      if (ok.second.my_flags) {
        unsigned long long one = 1;
        (void)hipMemcpy(ok.second.my_flags + 3, &one, sizeof(one),
                        hipMemcpyHostToDevice);
      }
    }
  }
}

// Lazy RCCL communicator for a process set (reference: nccl_operations.cc
// 87-131 — id bcast via the controller, lock-step on every global rank).
ncclComm_t EnsureComm(GlobalState& st, DeviceCtx& ctx, int32_t set_id) {
  {
    std::lock_guard<std::mutex> g(g_comms_mu);
    auto it = ctx.comms.find(set_id);
    if (it != ctx.comms.end()) return it->second;
  }
  auto& set = st.controller->process_set(set_id);
  int leader = set.ranks.empty() ? 0 : set.ranks[0];

  ncclUniqueId id;
  std::string payload;
  if (st.rank == leader) {
    RCCL_CHECK(ncclGetUniqueId(&id));
    payload.assign((const char*)&id, sizeof(id));
  }
  std::string idb = ExchangeUniqueId(st, leader, payload);
  if (idb.size() != sizeof(id))
    throw std::runtime_error("horovod_amd: RCCL uniqueId exchange failed");
  std::memcpy(&id, idb.data(), sizeof(id));

  c10::hip::HIPGuard guard(ctx.device);
  ncclComm_t comm = nullptr;
  RCCL_CHECK(ncclCommInitRank(&comm, (int)set.ranks.size(), id,
                              set.local_index(st.rank)));
  {
    std::lock_guard<std::mutex> g(g_comms_mu);
    ctx.comms[set_id] = comm;
  }
  g_rccl_used = true;
  EnsureWatchdog();
  HVD_LOG(INFO, "RCCL comm ready: process set %d, %d ranks, device %d",
          (int)set_id, (int)set.ranks.size(), ctx.device);
  return comm;
}

// Process sets whose RCCL comm bootstrap already ran on this rank (background
// thread only — every rank passes the same first-response for a set).
std::unordered_map<int32_t, bool> g_bootstrapped;
// Same for the one-shot shared-memory windows.
std::unordered_map<int32_t, bool> g_oneshot_bootstrapped;
// ...and for the hierarchical sub-comms.
std::unordered_map<int32_t, bool> g_hier_bootstrapped;

// ---- one-shot allreduce host side -----------------------------------------

// Crossover threshold is autotuner-adjustable at runtime (TUNE responses
// apply it rank-synchronously in response order, so eligibility stays
// consistent across ranks).  Staging capacity is fixed at the MAX so a
// raised threshold never overflows the IPC windows.
std::atomic<int64_t> g_oneshot_threshold{-1};

int64_t OneshotThresholdMax() {
  static int64_t v = [] {
    const char* e = std::getenv("HOROVOD_ONESHOT_THRESHOLD_MAX");
    return e ? (int64_t)atoll(e) : (int64_t)(8 << 20);
  }();
  return v;
}

int64_t OneshotThreshold() {
  int64_t v = g_oneshot_threshold.load(std::memory_order_relaxed);
  if (v < 0) {
    const char* e = std::getenv("HOROVOD_ONESHOT_THRESHOLD");
    v = e ? (int64_t)atoll(e) : (int64_t)(4 << 20);
    g_oneshot_threshold.store(v, std::memory_order_relaxed);
  }
  return std::min(v, OneshotThresholdMax());
}

bool OneshotEnabled() {
  static bool v = std::getenv("HOROVOD_ONESHOT_ALLREDUCE") != nullptr;
  return v;
}

// Eligibility is a pure function of the RESPONSE (not the local entries), so
// every global rank — member or relay — classifies identically and the
// star-frame lockstep holds.
bool OneshotEligible(const Response& resp, int n) {
  if (!OneshotEnabled() || resp.type != ResponseType::ALLREDUCE) return false;
  if (n < 2 || n > kOneshotMaxRanks) return false;
  switch (resp.dtype) {
    case DataType::HVD_FLOAT32:
    case DataType::HVD_FLOAT64:
    case DataType::HVD_FLOAT16:
    case DataType::HVD_BFLOAT16: break;
    default: return false;
  }
  switch (resp.reduce_op) {
    case ReduceOp::SUM:
    case ReduceOp::AVERAGE:
    case ReduceOp::MIN:
    case ReduceOp::MAX:
    case ReduceOp::PRODUCT: break;
    default: return false;
  }
  int64_t total = 0;
  const int64_t* p = resp.tensor_shapes.data();
  const int64_t* end = p + resp.tensor_shapes.size();
  while (p < end) {
    int64_t nd = *p++;
    int64_t numel = 1;
    for (int64_t i = 0; i < nd; ++i) numel *= *p++;
    total += AlignedElems(numel);
  }
  return total * (int64_t)DataTypeSize(resp.dtype) <= OneshotThreshold();
}

int OneshotOpCode(ReduceOp op) {
  switch (op) {
    case ReduceOp::MIN: return 1;
    case ReduceOp::MAX: return 2;
    case ReduceOp::PRODUCT: return 3;
    default: return 0;  // sum / average
  }
}

// ---- hierarchical / torus allreduce ---------------------------------------
// Two-level schedule for multi-node jobs (reference: NCCLHierarchicalAllreduce
// nccl_operations.cc:307-577 and the non-upstream NCCLTorusAllreduce 606-829):
// intra-node ncclReduceScatter -> cross-node ncclAllReduce on the shard (one
// GPU per node column) -> intra-node ncclAllGather.  All-RCCL (the reference
// hops through host MPI for the cross step; RCCL's own transports cover it).
// Enabled with HOROVOD_HIERARCHICAL_ALLREDUCE=1 when cross_size > 1; a
// single-node test can fake a 2x4 topology via the HOROVOD_LOCAL_* env.

bool HierEnabled() {
  static bool v = std::getenv("HOROVOD_HIERARCHICAL_ALLREDUCE") != nullptr;
  return v;
}

bool HierEligible(GlobalState& st, const Response& resp, int n_set) {
  if (!HierEnabled() || resp.type != ResponseType::ALLREDUCE) return false;
  if (n_set != st.size || st.local_size <= 1 || st.cross_size <= 1)
    return false;
  // contiguous rank layout (rank = cross_rank*local_size + local_rank) is
  // what the launchers produce; anything else falls back to flat
  return st.rank == st.cross_rank * st.local_size + st.local_rank;
}

// All-rank string exchange over the star (every global rank participates).
std::vector<std::string> ExchangeAllGather(GlobalState& st,
                                           const std::string& payload) {
  auto gathered = st.comm.Gather(payload);
  std::string joined;
  if (st.comm.is_root()) {
    for (auto& g : gathered) {
      uint32_t len = (uint32_t)g.size();
      joined.append((const char*)&len, sizeof(len));
      joined += g;
    }
  }
  joined = st.comm.Bcast(joined);
  std::vector<std::string> out;
  size_t pos = 0;
  while (pos + sizeof(uint32_t) <= joined.size()) {
    uint32_t len;
    std::memcpy(&len, joined.data() + pos, sizeof(len));
    pos += sizeof(len);
    out.push_back(joined.substr(pos, len));
    pos += len;
  }
  return out;
}

// Set up the shared-memory window for (ctx.device, set).  Lock-step: every
// global rank calls this (members allocate+open, non-members relay frames).
// Any failure anywhere degrades EVERY member to the RCCL path (a second
// star round agrees on the outcome) — never a mixed algorithm choice.
bool EnsureOneshot(GlobalState& st, DeviceCtx* ctx, int32_t set_id,
                   bool member) {
  OneshotCtx* os = nullptr;
  std::string payload;
  bool ok = true;
  if (member && ctx) {
    os = &ctx->oneshot[set_id];
    auto& set = st.controller->process_set(set_id);
    os->n = (int)set.ranks.size();
    os->li = set.local_index(st.rank);
    os->slot_bytes = OneshotThresholdMax();
    c10::hip::HIPGuard guard(ctx->device);
    hipIpcMemHandle_t hs{}, hf{};
    do {
      if (hipMalloc(&os->my_staging,
                    (size_t)(os->nslots * os->slot_bytes)) != hipSuccess) {
        ok = false;
        break;
      }
      if (hipExtMallocWithFlags((void**)&os->my_flags, 4096,
                                hipDeviceMallocFinegrained) != hipSuccess) {
        ok = false;
        break;
      }
      if (hipMemset(os->my_staging, 0,
                    (size_t)(os->nslots * os->slot_bytes)) != hipSuccess ||
          hipMemset(os->my_flags, 0, 4096) != hipSuccess) {
        ok = false;
        break;
      }
      if (hipIpcGetMemHandle(&hs, os->my_staging) != hipSuccess ||
          hipIpcGetMemHandle(&hf, os->my_flags) != hipSuccess) {
        ok = false;
        break;
      }
    } while (false);
    if (ok) {
      payload.assign((const char*)&hs, sizeof(hs));
      payload.append((const char*)&hf, sizeof(hf));
    }
  }
  auto all = ExchangeAllGather(st, payload);
  if (member && os && ok) {
    auto& set = st.controller->process_set(set_id);
    c10::hip::HIPGuard guard(ctx->device);
    for (int r = 0; r < os->n && ok; ++r) {
      int grank = set.ranks[r];
      if (grank == st.rank) {
        os->peer_staging[r] = os->my_staging;
        os->peer_flags[r] = os->my_flags;
        continue;
      }
      if (grank >= (int)all.size() ||
          all[grank].size() != 2 * sizeof(hipIpcMemHandle_t)) {
        ok = false;
        break;
      }
      hipIpcMemHandle_t hs{}, hf{};
      std::memcpy(&hs, all[grank].data(), sizeof(hs));
      std::memcpy(&hf, all[grank].data() + sizeof(hs), sizeof(hf));
      void *ps = nullptr, *pf = nullptr;
      if (hipIpcOpenMemHandle(&ps, hs, hipIpcMemLazyEnablePeerAccess) !=
              hipSuccess ||
          hipIpcOpenMemHandle(&pf, hf, hipIpcMemLazyEnablePeerAccess) !=
              hipSuccess) {
        ok = false;
        break;
      }
      os->peer_staging[r] = ps;
      os->peer_flags[r] = pf;
      os->opened.push_back(ps);
      os->opened.push_back(pf);
    }
  }
  // unanimous verdict round: one byte per rank ('1' ok / '0' failed)
  auto verdicts = ExchangeAllGather(st, member ? std::string(ok ? "1" : "0")
                                               : std::string("1"));
  bool all_ok = true;
  for (auto& v : verdicts)
    if (v == "0") all_ok = false;
  if (member && os) {
    os->ready = ok && all_ok;
    os->failed = !os->ready;
    if (os->failed)
      HVD_LOG(WARNING,
              "one-shot allreduce bootstrap failed for set %d; falling back "
              "to RCCL",
              (int)set_id);
    else
      HVD_LOG(INFO, "one-shot xGMI allreduce ready: set %d, %d ranks",
              (int)set_id, os->n);
  }
  return all_ok && (!member || ok);
}

at::Tensor& FusionBuffer(DeviceCtx& ctx, int64_t bytes) {
  if (!ctx.fusion_buffer.defined() || ctx.fusion_buffer.numel() < bytes) {
    c10::hip::HIPStreamGuard sg(ctx.stream);
    ctx.fusion_buffer = at::empty(
        {bytes}, at::TensorOptions().dtype(at::kByte).device(at::kCUDA, ctx.device));
  }
  return ctx.fusion_buffer;
}

void RecordStreamFor(const at::Tensor& t, const c10::hip::HIPStream& s) {
  if (t.defined() && !t.is_cpu())
    c10::hip::HIPCachingAllocator::recordStream(t.storage().data_ptr(), s);
}

// contiguous() with cross-stream ordering: the copy launches on the bg
// thread's current (default) stream, but consumption happens on the comm
// stream — fence it, or the RCCL call may read a half-written buffer.
at::Tensor ContigForComm(DeviceCtx& ctx, const at::Tensor& t) {
  at::Tensor c = t.contiguous();
  if (c.data_ptr() != t.data_ptr()) {
    hipEvent_t ev = AcquireEvent();
    HIP_CHECK(hipEventRecord(
        ev, c10::hip::getCurrentHIPStream(ctx.device).stream()));
    HIP_CHECK(hipStreamWaitEvent(ctx.stream.stream(), ev, 0));
    ReleaseEvent(ev);  // stream-wait snapshots the event; pool reuse is safe
    RecordStreamFor(c, ctx.stream);
  }
  return c;
}

void WaitReadyEvents(DeviceCtx& ctx, std::vector<TensorTableEntry>& entries,
                     std::vector<hipEvent_t>& ready) {
  bool any_per_tensor = false;
  for (auto& e : entries) {
    if (e.ready_event) {
      any_per_tensor = true;
      HIP_CHECK(hipStreamWaitEvent(ctx.stream.stream(), (hipEvent_t)e.ready_event, 0));
      ready.push_back((hipEvent_t)e.ready_event);
      e.ready_event = 0;
    }
    RecordStreamFor(e.tensor, ctx.stream);
    RecordStreamFor(e.output, ctx.stream);
  }
  if (!any_per_tensor && !entries.empty()) {
    // one producer-ordering event for the whole fused response: recorded on
    // the default (compute) stream, it covers every gradient already
    // enqueued by the training thread (see core.cc MakeEntry note)
    hipEvent_t ev = AcquireEvent();
    HIP_CHECK(hipEventRecord(
        ev, c10::hip::getCurrentHIPStream(ctx.device).stream()));
    HIP_CHECK(hipStreamWaitEvent(ctx.stream.stream(), ev, 0));
    ready.push_back(ev);
  }
}

void Finalize(DeviceCtx& ctx, std::vector<TensorTableEntry> entries,
              std::vector<hipEvent_t> ready, const char* activity,
              int64_t start_us, ncclComm_t comm = nullptr) {
  PendingOp op;
  op.done_event = AcquireEvent();
  HIP_CHECK(hipEventRecord(op.done_event, ctx.stream.stream()));
  op.entries = std::move(entries);
  op.ready_events = std::move(ready);
  op.device = ctx.device;
  op.activity = activity;
  op.start_us = start_us;
  op.comm = comm;
  EnsureFinalizer();
  {
    std::lock_guard<std::mutex> g(g_finalizer.mu);
    g_finalizer.queue.push_back(std::move(op));
  }
  g_finalizer.cv.notify_one();
}

// Pack entries (with prescale + dtype conversion) into the fusion buffer at
// 64-element-aligned offsets; returns total wire elements.
int64_t PackEntries(DeviceCtx& ctx, std::vector<TensorTableEntry>& entries,
                    DataType wire, bool unpack, char* base_override = nullptr) {
  int64_t wire_size = (int64_t)DataTypeSize(wire);
  int64_t total = 0;
  for (auto& e : entries) total += AlignedElems(e.tensor.numel());
  char* base = base_override;
  if (!base) {
    auto& buf = FusionBuffer(ctx, total * wire_size);
    base = (char*)buf.data_ptr();
  }

  CopyBatchArgs args;
  bool any_scale = false;
  int64_t off = 0;
  int src_dt = 0, dst_dt = 0;
  auto flush = [&](int s_dt, int d_dt) {
    if (args.count == 0) return;
    // fill the chip: ~2048 workgroups total regardless of batch size
    int bpc = std::min(256, std::max(8, 2048 / args.count));
    HIP_CHECK(BatchedCopyLaunch(args, s_dt, d_dt, any_scale, bpc,
                                ctx.stream.stream()));
    args.count = 0;
    any_scale = false;
  };
  for (auto& e : entries) {
    at::Tensor t = unpack ? e.output : e.tensor;
    if (!t.defined()) t = e.tensor;
    int this_src = unpack ? (int)wire : (int)DataTypeFromTorch(t.scalar_type());
    int this_dst = unpack ? (int)DataTypeFromTorch(t.scalar_type()) : (int)wire;
    if (args.count == kCopyBatchCapacity ||
        (args.count > 0 && (this_src != src_dt || this_dst != dst_dt)))
      flush(src_dt, dst_dt);
    src_dt = this_src;
    dst_dt = this_dst;
    double scale = unpack ? e.postscale : e.prescale;
    int i = args.count++;
    if (unpack) {
      args.src[i] = base + off * wire_size;
      args.dst[i] = t.data_ptr();
    } else {
      args.src[i] = t.data_ptr();
      args.dst[i] = base + off * wire_size;
    }
    args.numel[i] = (unsigned long long)t.numel();
    args.scale[i] = scale;
    if (scale != 1.0) any_scale = true;
    off += AlignedElems(t.numel());
  }
  flush(src_dt, dst_dt);
  return total;
}


// Create the local + cross RCCL comms for the hierarchical schedule.
// Lock-step over the star: every rank participates in one id-exchange
// round (local leaders contribute the local id, node-0 ranks the cross id
// for their column), then all ranks init local comms, then cross comms.
bool EnsureHier(GlobalState& st, DeviceCtx& ctx, int32_t set_id) {
  auto& h = ctx.hier[set_id];
  if (h.ready) return true;
  int ls = st.local_size, cs = st.cross_size, lr = st.local_rank,
      cr = st.cross_rank;
  ncclUniqueId lid{}, cid{};
  std::string payload(2 * sizeof(ncclUniqueId), '\0');
  if (lr == 0) {
    RCCL_CHECK(ncclGetUniqueId(&lid));
    std::memcpy(&payload[0], &lid, sizeof(lid));
  }
  if (cr == 0) {
    RCCL_CHECK(ncclGetUniqueId(&cid));
    std::memcpy(&payload[sizeof(lid)], &cid, sizeof(cid));
  }
  auto all = ExchangeAllGather(st, payload);
  int local_leader = cr * ls;          // local_rank 0 on my node
  int cross_leader = lr;               // my column's rank on node 0
  if (local_leader >= (int)all.size() || cross_leader >= (int)all.size() ||
      all[local_leader].size() < 2 * sizeof(ncclUniqueId) ||
      all[cross_leader].size() < 2 * sizeof(ncclUniqueId))
    return false;
  std::memcpy(&lid, all[local_leader].data(), sizeof(lid));
  std::memcpy(&cid, all[cross_leader].data() + sizeof(lid), sizeof(cid));
  c10::hip::HIPGuard guard(ctx.device);
  RCCL_CHECK(ncclCommInitRank(&h.local, ls, lid, lr));
  RCCL_CHECK(ncclCommInitRank(&h.cross, cs, cid, cr));
  h.ready = true;
  HVD_LOG(INFO, "hierarchical comms ready: %dx%d (local x cross)", ls, cs);
  return true;
}

// ---- Distributed VHDD Adasum (reference: adasum_gpu_operations.cc:44-120
// structure + adasum.h:195-345 FusedAllreduce) -----------------------------
//
// Vector-halving distance-doubling on the fused buffer: at level k each
// rank exchanges half of its current piece with partner li^2^k over RCCL
// point-to-point, computes per-tensor partial dot/norm products on its
// kept half (CDNA4 kernels), sums the scalars across the sharing group
// over the control star (3 doubles/tensor — tiny), and applies the
// projection-weighted combine in place.  O(total) memory and O(total)
// compute per rank vs the one-shot tree's n x both; the one-shot tree
// (below) remains the small-bucket option.
//
// Only valid for the GLOBAL process set: every global rank must relay the
// per-level scalar star rounds, and non-members of a subset op return
// before Execute's op switch.

int64_t AdasumOneshotThreshold() {
  static int64_t v = [] {
    const char* e = std::getenv("HOROVOD_ADASUM_ONESHOT_THRESHOLD");
    return e ? (int64_t)atoll(e) : (int64_t)(1 << 20);
  }();
  return v;
}

void ExecuteAdasumRSVHDD(GlobalState& st, DeviceCtx& ctx, Response& resp,
                         std::vector<TensorTableEntry>& entries,
                         ncclComm_t comm) {
  auto& set = st.controller->process_set(resp.process_set_id);
  const int n = (int)set.ranks.size();
  const int li = set.local_index(st.rank);
  DataType wire = resp.dtype;
  const int64_t wsz = (int64_t)DataTypeSize(wire);
  hipStream_t stream = ctx.stream.stream();
  auto wire_nccl = ToNccl(wire);

  const int64_t L = PackEntries(ctx, entries, wire, false);  // 64-aligned
  char* base = (char*)ctx.fusion_buffer.data_ptr();

  int p = 1;
  while (p * 2 <= n) p *= 2;
  int levels = 0;
  while ((1 << levels) < p) ++levels;

  // scratch for received pieces (<= L elems), device dots (3/tensor)
  if (!ctx.adasum_buffer.defined() || ctx.adasum_buffer.numel() < L * wsz) {
    c10::hip::HIPStreamGuard sg(ctx.stream);
    ctx.adasum_buffer = at::empty(
        {L * wsz},
        at::TensorOptions().dtype(at::kByte).device(at::kCUDA, ctx.device));
  }
  char* scratch = (char*)ctx.adasum_buffer.data_ptr();
  const int64_t T = (int64_t)entries.size();
  const int64_t ndots = T * 3;
  if (!ctx.dots_buffer.defined() || ctx.dots_buffer.numel() < ndots) {
    c10::hip::HIPStreamGuard sg(ctx.stream);
    ctx.dots_buffer = at::empty(
        {ndots},
        at::TensorOptions().dtype(at::kDouble).device(at::kCUDA, ctx.device));
  }
  double* ddots = ctx.dots_buffer.data_ptr<double>();

  // per-tensor REAL ranges in fused-element coordinates (padding excluded
  // from dot products — reference per-tensor coefficient isolation)
  std::vector<int64_t> offs(T), cnts(T);
  {
    int64_t off = 0;
    for (int64_t t = 0; t < T; ++t) {
      offs[t] = off;
      cnts[t] = entries[t].tensor.numel();
      off += AlignedElems(cnts[t]);
    }
  }

  // run the per-tensor kernels over the intersection of piece [S,E) with
  // each tensor's real range; a_is_mine controls canonical (a=lower-rank)
  // pointer order for dots, and which buffer ScaledAdd mutates
  auto run_frags = [&](int64_t S, int64_t E, bool mine_is_a, bool scaled_add,
                       int64_t piece_base /* scratch index of S */) {
    AdasumBatchArgs args;
    args.count = 0;
    int64_t dot_idx0 = -1;
    auto flush = [&](bool add) {
      if (args.count == 0) return;
      if (add)
        HIP_CHECK(AdasumScaledAddLaunch(args, (int)wire,
                                        ddots + dot_idx0 * 3, stream));
      else
        HIP_CHECK(AdasumDotsLaunch(args, (int)wire, ddots + dot_idx0 * 3,
                                   stream));
      args.count = 0;
      dot_idx0 = -1;
    };
    for (int64_t t = 0; t < T; ++t) {
      int64_t lo = std::max(S, offs[t]);
      int64_t hi = std::min(E, offs[t] + cnts[t]);
      char* mine = base + lo * wsz;
      char* theirs = scratch + (lo - S + piece_base) * wsz;
      // every tensor keeps a dots slot (group sums stay aligned), even when
      // this piece doesn't intersect it (numel 0)
      if (args.count == kCopyBatchCapacity) flush(scaled_add);
      if (dot_idx0 < 0) dot_idx0 = t;
      int k = args.count++;
      if (scaled_add) {
        // result goes into MY buffer regardless of role; the host swapped
        // na/nb for the higher rank so the kernel's coefficient order is
        // right for (a=mine, b=theirs)
        args.a[k] = mine;
        args.b[k] = theirs;
      } else {
        args.a[k] = mine_is_a ? (void*)mine : (void*)theirs;
        args.b[k] = mine_is_a ? (const void*)theirs : (const void*)mine;
      }
      args.numel[k] = (unsigned long long)std::max<int64_t>(0, hi - lo);
    }
    flush(scaled_add);
  };

  // star exchange of per-tensor partial dots; every global rank calls this
  // exactly `levels` times (active ranks with data, folded ranks empty)
  auto scalar_round = [&](std::vector<double>& partials, int group_lo,
                          int group_hi) {
    std::string payload;
    if (!partials.empty())
      payload.assign((const char*)partials.data(),
                     partials.size() * sizeof(double));
    auto all = ExchangeAllGather(st, payload);
    if (partials.empty()) return;
    std::fill(partials.begin(), partials.end(), 0.0);
    for (int r = group_lo; r < group_hi; ++r) {
      int grank = set.ranks[r];
      if (grank >= (int)all.size()) continue;
      auto& s = all[grank];
      if (s.size() != partials.size() * sizeof(double)) continue;
      const double* d = (const double*)s.data();
      for (size_t i = 0; i < partials.size(); ++i) partials[i] += d[i];
    }
  };

  // ---- fold the non-power-of-2 remainder: pair (i-p, i), local dots -----
  if (li >= p) {
    RCCL_CHECK(ncclGroupStart());
    RCCL_CHECK(ncclSend(base, L, wire_nccl, li - p, comm, stream));
    RCCL_CHECK(ncclGroupEnd());
  } else if (li + p < n) {
    RCCL_CHECK(ncclGroupStart());
    RCCL_CHECK(ncclRecv(scratch, L, wire_nccl, li + p, comm, stream));
    RCCL_CHECK(ncclGroupEnd());
    HIP_CHECK(hipMemsetAsync(ddots, 0, (size_t)ndots * sizeof(double),
                             stream));
    run_frags(0, L, /*mine_is_a=*/true, /*scaled_add=*/false, 0);
    run_frags(0, L, true, /*scaled_add=*/true, 0);
  }

  // ---- VHDD levels -------------------------------------------------------
  int64_t S = 0, len = L;
  std::vector<double> host_dots((size_t)ndots);
  for (int k = 0; k < levels; ++k) {
    int stride = 1 << k;
    if (li < p) {
      int partner = li ^ stride;
      bool lower = li < partner;
      int64_t half = len / 2;
      int64_t myS = lower ? S : S + half;
      int64_t thS = lower ? S + half : S;
      RCCL_CHECK(ncclGroupStart());
      RCCL_CHECK(ncclSend(base + thS * wsz, half, wire_nccl, partner, comm,
                          stream));
      RCCL_CHECK(ncclRecv(scratch, half, wire_nccl, partner, comm, stream));
      RCCL_CHECK(ncclGroupEnd());
      HIP_CHECK(hipMemsetAsync(ddots, 0, (size_t)ndots * sizeof(double),
                               stream));
      run_frags(myS, myS + half, lower, false, 0);
      HIP_CHECK(hipMemcpyAsync(host_dots.data(), ddots,
                               (size_t)ndots * sizeof(double),
                               hipMemcpyDeviceToHost, stream));
      SyncStreamBounded(stream);
      int group = 2 * stride;
      int g0 = (li / group) * group;
      scalar_round(host_dots, g0, g0 + group);
      if (!lower)
        for (int64_t t = 0; t < T; ++t)
          std::swap(host_dots[t * 3 + 1], host_dots[t * 3 + 2]);
      HIP_CHECK(hipMemcpyAsync(ddots, host_dots.data(),
                               (size_t)ndots * sizeof(double),
                               hipMemcpyHostToDevice, stream));
      run_frags(myS, myS + half, lower, true, 0);
      S = myS;
      len = half;
    } else {
      // folded rank: relay the scalar star round in lock-step
      std::vector<double> none;
      scalar_round(none, 0, 0);
    }
  }

  // ---- regather: grouped broadcast of the p final pieces -----------------
  const int64_t piece = L / p;
  RCCL_CHECK(ncclGroupStart());
  for (int r = 0; r < p; ++r) {
    int64_t Sr = 0;
    for (int k = 0; k < levels; ++k)
      if ((r >> k) & 1) Sr += L >> (k + 1);
    RCCL_CHECK(ncclBroadcast(base + Sr * wsz, base + Sr * wsz, piece,
                             wire_nccl, r, comm, stream));
  }
  RCCL_CHECK(ncclGroupEnd());

  for (auto& e : entries)
    if (!e.output.defined()) {
      e.output = at::empty_like(e.tensor);
      RecordStreamFor(e.output, ctx.stream);
    }
  PackEntries(ctx, entries, wire, true);
}

// One-shot Adasum over xGMI: allgather every rank's fused buffer (7 wide
// point-to-point links make this efficient on one MI355X node), then run the
// VHDD combine tree LOCALLY with the CDNA4 dot/scaled-add kernels — a
// bitwise-identical result on every rank with zero host synchronization.
// (Reference: AdasumGpuAllreduceOp, adasum_gpu_operations.cc:44-120, which
// does NCCL RS + host-MPI VHDD + NCCL AG; ours keeps everything on-device.)
void ExecuteAdasum(GlobalState& st, DeviceCtx& ctx, Response& resp,
                   std::vector<TensorTableEntry>& entries, ncclComm_t comm) {
  auto& set = st.controller->process_set(resp.process_set_id);
  int n = (int)set.ranks.size();
  if (n > 1 && set.ranks.size() == (size_t)st.size &&
      std::getenv("HOROVOD_ADASUM_ONESHOT") == nullptr) {
    // rank-consistent size check from the response (every member agrees)
    int64_t aligned = 0;
    const int64_t* ps = resp.tensor_shapes.data();
    const int64_t* pe = ps + resp.tensor_shapes.size();
    while (ps < pe) {
      int64_t nd = *ps++;
      int64_t numel = 1;
      for (int64_t i = 0; i < nd; ++i) numel *= *ps++;
      aligned += AlignedElems(numel);
    }
    if (aligned * (int64_t)DataTypeSize(resp.dtype) >=
        AdasumOneshotThreshold()) {
      ExecuteAdasumRSVHDD(st, ctx, resp, entries, comm);
      return;
    }
  }
  DataType wire = resp.dtype;
  int64_t wire_size = (int64_t)DataTypeSize(wire);
  hipStream_t stream = ctx.stream.stream();

  int64_t total = PackEntries(ctx, entries, wire, false);
  char* src_base = (char*)ctx.fusion_buffer.data_ptr();
  char* result_base = src_base;
  if (n > 1) {
    int64_t row_bytes = total * wire_size;
    if (!ctx.adasum_buffer.defined() ||
        ctx.adasum_buffer.numel() < (int64_t)n * row_bytes) {
      c10::hip::HIPStreamGuard sg(ctx.stream);
      ctx.adasum_buffer =
          at::empty({(int64_t)n * row_bytes},
                    at::TensorOptions().dtype(at::kByte).device(at::kCUDA, ctx.device));
    }
    char* ab = (char*)ctx.adasum_buffer.data_ptr();
    RCCL_CHECK(ncclAllGather(src_base, ab, total, ToNccl(wire), comm, stream));

    int64_t ndots = (int64_t)entries.size() * 3;
    if (!ctx.dots_buffer.defined() || ctx.dots_buffer.numel() < ndots) {
      c10::hip::HIPStreamGuard sg(ctx.stream);
      ctx.dots_buffer = at::empty(
          {ndots}, at::TensorOptions().dtype(at::kDouble).device(at::kCUDA, ctx.device));
    }
    double* dots = ctx.dots_buffer.data_ptr<double>();

    // per-tensor offsets within a row (64-elem aligned, matching PackEntries)
    std::vector<int64_t> offs, cnts;
    int64_t off = 0;
    for (auto& e : entries) {
      offs.push_back(off);
      cnts.push_back(e.tensor.numel());
      off += AlignedElems(e.tensor.numel());
    }
    auto combine = [&](int i, int j) {
      char* rowa = ab + (int64_t)i * row_bytes;
      char* rowb = ab + (int64_t)j * row_bytes;
      HIP_CHECK(hipMemsetAsync(dots, 0, (size_t)ndots * sizeof(double), stream));
      for (size_t start = 0; start < entries.size(); start += kCopyBatchCapacity) {
        AdasumBatchArgs a;
        a.count = (int)std::min<size_t>(kCopyBatchCapacity, entries.size() - start);
        for (int k = 0; k < a.count; ++k) {
          a.a[k] = rowa + offs[start + k] * wire_size;
          a.b[k] = rowb + offs[start + k] * wire_size;
          a.numel[k] = (unsigned long long)cnts[start + k];
        }
        HIP_CHECK(AdasumDotsLaunch(a, (int)wire, dots + start * 3, stream));
      }
      for (size_t start = 0; start < entries.size(); start += kCopyBatchCapacity) {
        AdasumBatchArgs a;
        a.count = (int)std::min<size_t>(kCopyBatchCapacity, entries.size() - start);
        for (int k = 0; k < a.count; ++k) {
          a.a[k] = rowa + offs[start + k] * wire_size;
          a.b[k] = rowb + offs[start + k] * wire_size;
          a.numel[k] = (unsigned long long)cnts[start + k];
        }
        HIP_CHECK(AdasumScaledAddLaunch(a, (int)wire, dots + start * 3, stream));
      }
    };
    // VHDD tree (matches the CPU/golden ordering in core.cc): fold the
    // non-power-of-2 remainder, then pairwise distance-doubling.
    int p = 1;
    while (p * 2 <= n) p *= 2;
    for (int i = p; i < n; ++i) combine(i - p, i);
    for (int stride = 1; stride < p; stride *= 2)
      for (int i = 0; i + stride < p; i += 2 * stride) combine(i, i + stride);
    result_base = ab;  // row 0
  }
  for (auto& e : entries)
    if (!e.output.defined()) {
      e.output = at::empty_like(e.tensor);
      RecordStreamFor(e.output, ctx.stream);
    }
  PackEntries(ctx, entries, wire, true, result_base);
}

}  // namespace

uintptr_t RecordReadyEvent(int device) {
  hipEvent_t ev = AcquireEvent();
  HIP_CHECK(hipEventRecord(ev, c10::hip::getCurrentHIPStream(device).stream()));
  return (uintptr_t)ev;
}

bool RcclUsed() { return g_rccl_used; }

void Execute(GlobalState& st, Response& resp,
             std::vector<TensorTableEntry>& entries) {
  if (g_comm_failed) {
    // fail fast until elastic re-init replaces the comms — issuing RCCL
    // calls on an aborted comm would hang or corrupt the stream
    Status s = Status::Aborted(
        "RCCL communicator aborted (peer failure or async error)");
    for (auto& e : entries)
      if (e.callback) e.callback(s, e);
    return;
  }
  auto& set = st.controller->process_set(resp.process_set_id);
  const int32_t sid = resp.process_set_id;
  // Eligibility is response-derived, so members and relay-only ranks agree
  // on which bootstrap (one-shot window vs RCCL comm) this response needs.
  bool oneshot_resp = OneshotEligible(resp, (int)set.ranks.size());
  auto rccl_bootstrap = [&] {
    // Every global rank reaches this point on the set's first RCCL response
    // (responses are broadcast in a fixed order), so members can init the
    // RCCL comm while non-members relay the uniqueId frames in lock-step.
    if (!entries.empty()) {
      EnsureComm(st, GetCtx(entries[0].device), sid);
    } else {
      ExchangeUniqueId(st, set.ranks.empty() ? 0 : set.ranks[0], "");
    }
    g_bootstrapped[sid] = true;
  };
  if (oneshot_resp && !g_oneshot_bootstrapped.count(sid)) {
    bool ok = EnsureOneshot(st, entries.empty() ? nullptr
                                                : &GetCtx(entries[0].device),
                            sid, !entries.empty());
    g_oneshot_bootstrapped[sid] = true;
    // degraded verdict is GLOBAL: every rank (relay included) runs the RCCL
    // bootstrap now so members can fall back without desyncing the star
    if (!ok && !g_bootstrapped.count(sid)) rccl_bootstrap();
  }
  if (!oneshot_resp && !g_bootstrapped.count(sid)) rccl_bootstrap();
  // hierarchical comms piggyback their one id-exchange round on the first
  // eligible response (global set => every rank is a member and reaches
  // this in lock-step)
  bool hier_resp = !oneshot_resp && HierEligible(st, resp, (int)set.ranks.size());
  if (hier_resp && !g_hier_bootstrapped.count(sid)) {
    if (!entries.empty()) {
      EnsureHier(st, GetCtx(entries[0].device), sid);
    } else {
      ExchangeAllGather(st, "");  // relay (unreachable for global sets)
    }
    g_hier_bootstrapped[sid] = true;
  }
  if (entries.empty()) return;  // relay-only rank (not a member)
  if (entries.size() != resp.names.size())
    throw std::runtime_error(
        "horovod_amd: fused response entry/name count mismatch (" +
        std::to_string(entries.size()) + " vs " +
        std::to_string(resp.names.size()) +
        ") — fusion layout would diverge across ranks");
  int device = entries[0].device;
  auto& ctx = GetCtx(device);
  OneshotCtx* os = nullptr;
  if (oneshot_resp) {
    auto oit = ctx.oneshot.find(sid);
    if (oit != ctx.oneshot.end() && oit->second.ready) os = &oit->second;
  }
  // one-shot ops never touch RCCL; every other path (and the degraded
  // fallback) uses the communicator created by the lock-step bootstrap
  ncclComm_t comm = os ? nullptr : EnsureComm(st, ctx, sid);
  c10::hip::HIPGuard dguard(device);
  hipStream_t stream = ctx.stream.stream();
  int n = (int)set.ranks.size();
  int li = set.local_index(st.rank);
  DataType wire = resp.dtype;
  int64_t wire_size = (int64_t)DataTypeSize(wire);
  auto wire_nccl = ToNccl(wire);
  auto tl0 = GetTimeline(st);
  int64_t t_start = tl0 ? tl0->NowUs() : 0;

  std::vector<hipEvent_t> ready;
  WaitReadyEvents(ctx, entries, ready);

  // debug bisection knob: complete without any GPU work (timing diagnosis)
  static const bool noop_exec = std::getenv("HVD_DEBUG_NOOP_EXEC") != nullptr;
  if (noop_exec) {
    for (auto& e : entries)
      if (!e.output.defined()) e.output = e.tensor;
    Finalize(ctx, std::move(entries), std::move(ready), "NOOP", t_start);
    return;
  }

  // roctx range so external profilers (rocprofv3 --marker-trace) see the op
  // (reference: SharedNvtxOpRange, nvtx_op_range.{cc,h}); disable via
  // HOROVOD_DISABLE_ROCTX_RANGES.
  static const bool roctx_on = std::getenv("HOROVOD_DISABLE_ROCTX_RANGES") == nullptr;
  if (roctx_on)
    roctxRangePushA((std::string("hvd.") + entries[0].name).c_str());

  const char* activity = "RCCL_OP";
  switch (resp.type) {
    case ResponseType::ALLREDUCE: {
      if (os) {
        // one-shot over xGMI: pack into my shared slot, peers read all n
        // slots directly and reduce locally (oneshot.hip protocol)
        activity = "ONESHOT_ALLREDUCE";
        os->seq++;
        int slot = (int)(os->seq % (unsigned long long)os->nslots);
        OneshotDeviceArgs a{};
        for (int r = 0; r < os->n; ++r) {
          a.staging[r] =
              (char*)os->peer_staging[r] + (int64_t)slot * os->slot_bytes;
          a.flags[r] = os->peer_flags[r];
        }
        a.n = os->n;
        a.li = os->li;
        if (os->seq > (unsigned long long)os->nslots)
          HIP_CHECK(OneshotWaitConsumedLaunch(a, os->seq - os->nslots,
                                              stream));
        int64_t total = PackEntries(
            ctx, entries, wire, false,
            os->my_staging + (int64_t)slot * os->slot_bytes);
        auto& fb = FusionBuffer(ctx, total * wire_size);
        HIP_CHECK(OneshotReduceLaunch(
            a, os->seq, (int)wire, (unsigned long long)(total * wire_size),
            fb.data_ptr(), OneshotOpCode(resp.reduce_op), stream));
        for (auto& e : entries)
          if (!e.output.defined()) {
            e.output = at::empty_like(e.tensor);
            RecordStreamFor(e.output, ctx.stream);
          }
        PackEntries(ctx, entries, wire, true);
        break;
      }
      DeviceCtx::HierCtx* hc = nullptr;
      if (hier_resp) {
        auto hit = ctx.hier.find(sid);
        if (hit != ctx.hier.end() && hit->second.ready) hc = &hit->second;
      }
      if (hc) {
        // two-level: local RS -> cross AR on the shard -> local AG
        activity = "HIER_ALLREDUCE";
        int64_t total = PackEntries(ctx, entries, wire, false);
        char* base = (char*)ctx.fusion_buffer.data_ptr();
        int ls = st.local_size, lr = st.local_rank;
        if (total > 0 && total % ls == 0) {
          int64_t shard = total / ls;
          char* mine = base + (int64_t)lr * shard * wire_size;
          RCCL_CHECK(ncclReduceScatter(base, mine, shard, wire_nccl,
                                       ToNcclOp(resp.reduce_op), hc->local,
                                       stream));
          RCCL_CHECK(ncclAllReduce(mine, mine, shard, wire_nccl,
                                   ToNcclOp(resp.reduce_op), hc->cross,
                                   stream));
          RCCL_CHECK(ncclAllGather(mine, base, shard, wire_nccl, hc->local,
                                   stream));
        } else {
          // fused length not divisible by local_size: flat fallback
          RCCL_CHECK(ncclAllReduce(base, base, total, wire_nccl,
                                   ToNcclOp(resp.reduce_op), comm, stream));
        }
        for (auto& e : entries)
          if (!e.output.defined()) {
            e.output = at::empty_like(e.tensor);
            RecordStreamFor(e.output, ctx.stream);
          }
        PackEntries(ctx, entries, wire, true);
        break;
      }
      activity = "RCCL_ALLREDUCE";
      // dense (possibly permuted, e.g. channels_last) tensors are raw-
      // copyable: identical layout on every rank, so RCCL/pack operate in
      // memory order consistently.
      bool direct =
          entries.size() == 1 && entries[0].prescale == 1.0 &&
          entries[0].postscale == 1.0 &&
          DataTypeFromTorch(entries[0].tensor.scalar_type()) == wire &&
          entries[0].tensor.is_non_overlapping_and_dense() &&
          entries[0].output.defined() &&
          entries[0].output.strides() == entries[0].tensor.strides();
      if (direct) {
        auto& e = entries[0];
        RCCL_CHECK(ncclAllReduce(e.tensor.data_ptr(), e.output.data_ptr(),
                                 e.tensor.numel(), wire_nccl,
                                 ToNcclOp(resp.reduce_op), comm, stream));
      } else {
        int64_t total = PackEntries(ctx, entries, wire, false);
        RCCL_CHECK(ncclAllReduce(ctx.fusion_buffer.data_ptr(),
                                 ctx.fusion_buffer.data_ptr(), total, wire_nccl,
                                 ToNcclOp(resp.reduce_op), comm, stream));
        for (auto& e : entries)
          if (!e.output.defined()) {
            // allocate on the DEFAULT stream (the consumer side) and record
            // the comm-stream use: frees then correctly fence both streams
            e.output = at::empty_like(e.tensor);
            RecordStreamFor(e.output, ctx.stream);
          }
        PackEntries(ctx, entries, wire, true);
      }
      break;
    }
    case ResponseType::ADASUM: {
      activity = "ADASUM";
      ExecuteAdasum(st, ctx, resp, entries, comm);
      break;
    }
    case ResponseType::BROADCAST: {
      activity = "RCCL_BCAST";
      auto& e = entries[0];
      int root_li = set.local_index(resp.root_rank);
      at::Tensor in = e.tensor.is_non_overlapping_and_dense()
                          ? e.tensor
                          : ContigForComm(ctx, e.tensor);
      if (!e.output.defined()) e.output = in;
      RCCL_CHECK(ncclBroadcast(in.data_ptr(), e.output.data_ptr(), in.numel(),
                               wire_nccl, root_li, comm, stream));
      break;
    }
    case ResponseType::ALLGATHER: {
      activity = "RCCL_ALLGATHER";
      auto& e = entries[0];
      at::Tensor in = ContigForComm(ctx, e.tensor);
      int64_t row_elems = e.tensor.numel();
      if (e.tensor.dim() > 0 && e.tensor.size(0) > 0)
        row_elems = e.tensor.numel() / e.tensor.size(0);
      int64_t total0 = 0;
      bool same = true;
      for (int r = 0; r < n; ++r) {
        total0 += resp.tensor_sizes[r];
        if (resp.tensor_sizes[r] != resp.tensor_sizes[0]) same = false;
      }
      std::vector<int64_t> out_shape(e.tensor.sizes().begin(),
                                     e.tensor.sizes().end());
      if (out_shape.empty()) out_shape = {total0};
      else out_shape[0] = total0;
      e.output = at::empty(out_shape, e.tensor.options());
      RecordStreamFor(e.output, ctx.stream);
      if (same && total0 > 0) {
        RCCL_CHECK(ncclAllGather(in.data_ptr(), e.output.data_ptr(),
                                 resp.tensor_sizes[0] * row_elems, wire_nccl, comm,
                                 stream));
      } else {
        // v-variant via grouped broadcasts (reference: nccl_operations.cc
        // 1082-1101)
        RCCL_CHECK(ncclGroupStart());
        char* out_base = (char*)e.output.data_ptr();
        int64_t off = 0;
        for (int r = 0; r < n; ++r) {
          int64_t cnt = resp.tensor_sizes[r] * row_elems;
          if (cnt > 0) {
            RCCL_CHECK(ncclBroadcast(
                r == li ? in.data_ptr() : (void*)(out_base + off * wire_size),
                out_base + off * wire_size, cnt, wire_nccl, r, comm, stream));
          }
          off += cnt;
        }
        RCCL_CHECK(ncclGroupEnd());
      }
      // per-rank first-dim sizes: lets the autograd backward compute its
      // slice offset without a dims-allgather every backward pass
      e.received_splits = at::empty({n}, at::kLong);
      auto* gs = e.received_splits.data_ptr<int64_t>();
      for (int r = 0; r < n; ++r) gs[r] = resp.tensor_sizes[r];
      break;
    }
    case ResponseType::ALLTOALL: {
      activity = "RCCL_ALLTOALL";
      auto& e = entries[0];
      at::Tensor in = ContigForComm(ctx, e.tensor);
      int64_t row_elems = 1;
      for (int d = 1; d < e.tensor.dim(); ++d) row_elems *= e.tensor.size(d);
      // splits matrix: row i = sender i's splits
      int64_t recv_rows = 0;
      for (int r = 0; r < n; ++r) recv_rows += resp.tensor_sizes[(size_t)r * n + li];
      std::vector<int64_t> out_shape(e.tensor.sizes().begin(),
                                     e.tensor.sizes().end());
      if (out_shape.empty()) out_shape = {recv_rows};
      else out_shape[0] = recv_rows;
      e.output = at::empty(out_shape, e.tensor.options());
      RecordStreamFor(e.output, ctx.stream);
      char* in_base = (char*)in.data_ptr();
      char* out_base = (char*)e.output.data_ptr();
      RCCL_CHECK(ncclGroupStart());
      int64_t send_off = 0, recv_off = 0;
      for (int r = 0; r < n; ++r) {
        int64_t s_rows = resp.tensor_sizes[(size_t)li * n + r];
        int64_t r_rows = resp.tensor_sizes[(size_t)r * n + li];
        if (s_rows > 0)
          RCCL_CHECK(ncclSend(in_base + send_off * row_elems * wire_size,
                              s_rows * row_elems, wire_nccl, r, comm, stream));
        if (r_rows > 0)
          RCCL_CHECK(ncclRecv(out_base + recv_off * row_elems * wire_size,
                              r_rows * row_elems, wire_nccl, r, comm, stream));
        send_off += s_rows;
        recv_off += r_rows;
      }
      RCCL_CHECK(ncclGroupEnd());
      e.received_splits = at::empty({n}, at::kLong);
      auto* rs = e.received_splits.data_ptr<int64_t>();
      for (int r = 0; r < n; ++r) rs[r] = resp.tensor_sizes[(size_t)r * n + li];
      break;
    }
    case ResponseType::REDUCESCATTER: {
      activity = "RCCL_REDUCESCATTER";
      auto& e = entries[0];
      at::Tensor in = ContigForComm(ctx, e.tensor);
      if (e.prescale != 1.0) {
        // scale into the fusion buffer before the RCCL call so prescale
        // matches the CPU path (FlatPrescaled) — previously silently dropped
        auto& buf = FusionBuffer(ctx, in.numel() * wire_size);
        CopyBatchArgs pargs;
        pargs.count = 1;
        pargs.src[0] = in.data_ptr();
        pargs.dst[0] = buf.data_ptr();
        pargs.numel[0] = (unsigned long long)in.numel();
        pargs.scale[0] = e.prescale;
        HIP_CHECK(BatchedCopyLaunch(pargs, (int)wire, (int)wire, true, 256,
                                    stream));
        in = buf;  // byte buffer: only data_ptr is used below
      }
      int64_t first = e.tensor.dim() > 0 ? e.tensor.size(0) : 1;
      int64_t row_elems = first > 0 ? e.tensor.numel() / first : 0;
      int64_t base_rows = first / n, rem = first % n;
      int64_t my_rows = base_rows + (li < rem ? 1 : 0);
      std::vector<int64_t> out_shape(e.tensor.sizes().begin(),
                                     e.tensor.sizes().end());
      if (out_shape.empty()) out_shape = {my_rows};
      else out_shape[0] = my_rows;
      e.output = at::empty(out_shape, e.tensor.options());
      RecordStreamFor(e.output, ctx.stream);
      if (rem == 0 && first > 0) {
        RCCL_CHECK(ncclReduceScatter(in.data_ptr(), e.output.data_ptr(),
                                     base_rows * row_elems, wire_nccl,
                                     ToNcclOp(resp.reduce_op), comm, stream));
      } else {
        // v-variant via grouped reduces (reference: nccl_operations.cc
        // 1294-1316)
        char* in_base = (char*)in.data_ptr();
        RCCL_CHECK(ncclGroupStart());
        int64_t off = 0;
        for (int r = 0; r < n; ++r) {
          int64_t rows = base_rows + (r < rem ? 1 : 0);
          if (rows > 0)
            RCCL_CHECK(ncclReduce(in_base + off * row_elems * wire_size,
                                  e.output.data_ptr(), rows * row_elems, wire_nccl,
                                  ToNcclOp(resp.reduce_op), r, comm, stream));
          off += rows;
        }
        RCCL_CHECK(ncclGroupEnd());
      }
      if (e.postscale != 1.0) {
        CopyBatchArgs args;
        args.count = 1;
        args.src[0] = e.output.data_ptr();
        args.dst[0] = e.output.data_ptr();
        args.numel[0] = (unsigned long long)e.output.numel();
        args.scale[0] = e.postscale;
        HIP_CHECK(BatchedCopyLaunch(args, (int)wire, (int)wire, true, 256, stream));
      }
      break;
    }
    default:
      throw std::runtime_error("horovod_amd: bad GPU response type");
  }

  if (roctx_on) roctxRangePop();
  Finalize(ctx, std::move(entries), std::move(ready), activity, t_start, comm);
}

void AbortComms(const std::string& why) {
  std::lock_guard<std::mutex> gc(g_ctx_mu);
  std::lock_guard<std::mutex> g(g_comms_mu);
  AbortAllCommsLocked(why.c_str());
}

bool CommsFailed() { return g_comm_failed; }

void SetOneshotThreshold(int64_t bytes) {
  g_oneshot_threshold.store(bytes, std::memory_order_relaxed);
}

void AdasumCombine(std::vector<at::Tensor>& a, std::vector<at::Tensor>& b) {
  if (a.empty()) return;
  int device = (int)a[0].get_device();
  c10::hip::HIPGuard guard(device);
  hipStream_t stream = c10::hip::getCurrentHIPStream(device).stream();
  at::Tensor dots_t = at::zeros(
      {(int64_t)a.size() * 3},
      at::TensorOptions().dtype(at::kDouble).device(at::kCUDA, device));
  double* dots = dots_t.data_ptr<double>();
  int dt = (int)DataTypeFromTorch(a[0].scalar_type());
  for (size_t start = 0; start < a.size(); start += kCopyBatchCapacity) {
    AdasumBatchArgs args;
    args.count = (int)std::min<size_t>(kCopyBatchCapacity, a.size() - start);
    for (int k = 0; k < args.count; ++k) {
      args.a[k] = a[start + k].data_ptr();
      args.b[k] = b[start + k].data_ptr();
      args.numel[k] = (unsigned long long)a[start + k].numel();
    }
    HIP_CHECK(AdasumDotsLaunch(args, dt, dots + start * 3, stream));
  }
  for (size_t start = 0; start < a.size(); start += kCopyBatchCapacity) {
    AdasumBatchArgs args;
    args.count = (int)std::min<size_t>(kCopyBatchCapacity, a.size() - start);
    for (int k = 0; k < args.count; ++k) {
      args.a[k] = a[start + k].data_ptr();
      args.b[k] = b[start + k].data_ptr();
      args.numel[k] = (unsigned long long)a[start + k].numel();
    }
    HIP_CHECK(AdasumScaledAddLaunch(args, dt, dots + start * 3, stream));
  }
}

void FusedSgdStep(std::vector<at::Tensor>& params,
                  std::vector<at::Tensor>& grads,
                  std::vector<at::Tensor>& momenta, double lr, double momentum,
                  double weight_decay, double dampening, bool nesterov) {
  if (params.empty()) return;
  int device = (int)params[0].get_device();
  c10::hip::HIPGuard guard(device);
  hipStream_t stream = c10::hip::getCurrentHIPStream(device).stream();
  SgdBatchArgs args;
  auto flush = [&] {
    if (args.count == 0) return;
    HIP_CHECK(FusedSgdLaunch(args, (float)lr, (float)momentum,
                             (float)weight_decay, (float)dampening, nesterov,
                             stream));
    args.count = 0;
  };
  for (size_t i = 0; i < params.size(); ++i) {
    if (args.count == kCopyBatchCapacity) flush();
    int k = args.count++;
    args.params[k] = params[i].data_ptr();
    args.grads[k] = grads[i].data_ptr();
    args.momenta[k] = momenta.empty() ? nullptr : momenta[i].data_ptr();
    args.numel[k] = (unsigned long long)params[i].numel();
  }
  flush();
}

void FusedAdamwStep(std::vector<at::Tensor>& params,
                    std::vector<at::Tensor>& grads,
                    std::vector<at::Tensor>& exp_avgs,
                    std::vector<at::Tensor>& exp_avg_sqs, double lr,
                    double beta1, double beta2, double eps,
                    double weight_decay, int64_t step) {
  if (params.empty()) return;
  int device = (int)params[0].get_device();
  c10::hip::HIPGuard guard(device);
  hipStream_t stream = c10::hip::getCurrentHIPStream(device).stream();
  AdamwBatchArgs args;
  auto flush = [&] {
    if (args.count == 0) return;
    HIP_CHECK(FusedAdamwLaunch(args, (float)lr, (float)beta1, (float)beta2,
                               (float)eps, (float)weight_decay, step,
                               stream));
    args.count = 0;
  };
  for (size_t i = 0; i < params.size(); ++i) {
    if (args.count == kCopyBatchCapacity) flush();
    int k = args.count++;
    args.params[k] = params[i].data_ptr();
    args.grads[k] = grads[i].data_ptr();
    args.exp_avg[k] = exp_avgs[i].data_ptr();
    args.exp_avg_sq[k] = exp_avg_sqs[i].data_ptr();
    args.numel[k] = (unsigned long long)params[i].numel();
  }
  flush();
}

// ---- Fused BN(+Add)+ReLU host side ----------------------------------------
// Returns {y, save_mean, save_invstd}.  Small per-channel math ([C] tensors)
// runs through ATen; the big passes are the CDNA4 kernels.
std::vector<at::Tensor> FusedBnReluForward(at::Tensor x, at::Tensor residual,
                                           at::Tensor gamma, at::Tensor beta,
                                           at::Tensor running_mean,
                                           at::Tensor running_var,
                                           double momentum, double eps) {
  TORCH_CHECK(x.dim() == 4, "fused BN expects NCHW logical 4-D input");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "fused BN requires channels_last layout");
  int64_t C = x.size(1);
  TORCH_CHECK(C % 8 == 0 && C <= 4096, "fused BN requires C % 8 == 0, C <= 4096");
  int device = (int)x.get_device();
  c10::hip::HIPGuard guard(device);
  hipStream_t stream = c10::hip::getCurrentHIPStream(device).stream();
  int64_t total = x.numel();
  int64_t count = total / C;
  int dt = (int)DataTypeFromTorch(x.scalar_type());

  auto fopts = at::TensorOptions().dtype(at::kFloat).device(x.device());
  // one workspace, zero ATen math on [C] tensors: the per-channel epilogue
  // (mean/invstd + running-stat update) runs in bn_finalize_k — the
  // previous ~10 tiny ATen launches per BN layer cost 3.5 ms/step of pure
  // launch overhead at batch 64 (53 layers, fwd+bwd)
  const int64_t bank_elems = (int64_t)kBnBanks * 2 * C;
  at::Tensor ws = at::empty({bank_elems + 2 * C}, fopts);
  float* wsp = ws.data_ptr<float>();
  float* mean_p = wsp + bank_elems;
  float* invstd_p = mean_p + C;
  HIP_CHECK(hipMemsetAsync(wsp, 0, (size_t)bank_elems * sizeof(float),
                           stream));
  HIP_CHECK(BnStatsLaunch(x.data_ptr(), total, (int)C, dt, wsp, nullptr,
                          stream));
  HIP_CHECK(BnBankReduceLaunch(wsp, (int)C, stream));
  bool stats_f32 = running_mean.defined() &&
                   running_mean.scalar_type() == at::kFloat &&
                   running_mean.is_contiguous() &&
                   running_var.scalar_type() == at::kFloat &&
                   running_var.is_contiguous();
  HIP_CHECK(BnFinalizeLaunch(
      wsp, mean_p, invstd_p,
      stats_f32 ? running_mean.data_ptr<float>() : nullptr,
      stats_f32 ? running_var.data_ptr<float>() : nullptr, count,
      (float)momentum, (float)eps, (int)C, stream));
  at::Tensor mean = ws.narrow(0, bank_elems, C);
  at::Tensor invstd = ws.narrow(0, bank_elems + C, C);
  if (!stats_f32 && running_mean.defined()) {
    // rare non-fp32 running stats: fall back to ATen for the update only
    at::Tensor var = at::pow(invstd, -2) - eps;
    running_mean.mul_(1 - momentum).add_(mean.to(running_mean.scalar_type()),
                                         momentum);
    double ub = count > 1 ? (double)count / (count - 1) : 1.0;
    running_var.mul_(1 - momentum)
        .add_((var * ub).to(running_var.scalar_type()), momentum);
  }
  at::Tensor gamma_f =
      gamma.scalar_type() == at::kFloat && gamma.is_contiguous()
          ? gamma
          : gamma.to(at::kFloat).contiguous();
  at::Tensor beta_f =
      beta.scalar_type() == at::kFloat && beta.is_contiguous()
          ? beta
          : beta.to(at::kFloat).contiguous();
  at::Tensor y = at::empty_like(x);
  HIP_CHECK(BnApplyReluLaunch(
      x.data_ptr(), residual.defined() ? residual.data_ptr() : nullptr,
      y.data_ptr(), mean_p, invstd_p, gamma_f.data_ptr<float>(),
      beta_f.data_ptr<float>(), total, (int)C, dt, stream));
  return {y, mean, invstd};
}

// Returns {dx, dgamma, dbeta[, dresidual]}.
std::vector<at::Tensor> FusedBnReluBackward(at::Tensor x, at::Tensor y,
                                            at::Tensor dy, at::Tensor mean,
                                            at::Tensor invstd, at::Tensor gamma,
                                            bool need_residual_grad) {
  int device = (int)x.get_device();
  c10::hip::HIPGuard guard(device);
  hipStream_t stream = c10::hip::getCurrentHIPStream(device).stream();
  int64_t C = x.size(1);
  int64_t total = x.numel();
  int64_t count = total / C;
  int dt = (int)DataTypeFromTorch(x.scalar_type());
  if (!dy.is_contiguous(at::MemoryFormat::ChannelsLast))
    dy = dy.contiguous(at::MemoryFormat::ChannelsLast);

  auto fopts = at::TensorOptions().dtype(at::kFloat).device(x.device());
  at::Tensor sums = at::empty({(int64_t)kBnBanks * 2 * C}, fopts);
  float* bankp = sums.data_ptr<float>();
  HIP_CHECK(hipMemsetAsync(bankp, 0,
                           (size_t)kBnBanks * 2 * C * sizeof(float), stream));
  at::Tensor dres;
  if (need_residual_grad) dres = at::empty_like(x);
  HIP_CHECK(BnBwdStatsLaunch(x.data_ptr(), y.data_ptr(), dy.data_ptr(),
                             need_residual_grad ? dres.data_ptr() : nullptr,
                             mean.data_ptr<float>(), invstd.data_ptr<float>(),
                             total, (int)C, dt, bankp, nullptr, stream));
  HIP_CHECK(BnBankReduceLaunch(bankp, (int)C, stream));
  at::Tensor gamma_f =
      gamma.scalar_type() == at::kFloat && gamma.is_contiguous()
          ? gamma
          : gamma.to(at::kFloat).contiguous();
  at::Tensor dx = at::empty_like(x);
  HIP_CHECK(BnBwdApplyLaunch(x.data_ptr(), y.data_ptr(), dy.data_ptr(),
                             dx.data_ptr(), mean.data_ptr<float>(),
                             invstd.data_ptr<float>(), gamma_f.data_ptr<float>(),
                             bankp, bankp + C, total, (int)C, dt,
                             (float)(1.0 / count), stream));
  at::Tensor sum_g = sums.narrow(0, 0, C), sum_gx = sums.narrow(0, C, C);
  at::Tensor dbeta = gamma.scalar_type() == at::kFloat
                         ? sum_g
                         : sum_g.to(gamma.scalar_type());
  at::Tensor dgamma = gamma.scalar_type() == at::kFloat
                          ? sum_gx
                          : sum_gx.to(gamma.scalar_type());
  std::vector<at::Tensor> out{dx, dgamma, dbeta};
  if (need_residual_grad) out.push_back(dres);
  return out;
}

void WaitAllPending() {
  while (true) {
    {
      std::lock_guard<std::mutex> g(g_finalizer.mu);
      if (g_finalizer.queue.empty()) break;
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(1));
  }
}

void Shutdown() {
  WaitAllPending();
  if (g_watchdog.started) {
    g_watchdog.stop = true;
    if (g_watchdog.thread.joinable()) g_watchdog.thread.join();
    g_watchdog.started = false;
  }
  {
    std::lock_guard<std::mutex> g(g_finalizer.mu);
    g_finalizer.stop = true;
  }
  g_finalizer.cv.notify_all();
  if (g_finalizer.started && g_finalizer.thread.joinable()) g_finalizer.thread.join();
  g_finalizer.started = false;
  g_finalizer.stop = false;
  {
    std::lock_guard<std::mutex> g(g_ctx_mu);
    std::lock_guard<std::mutex> g2(g_comms_mu);
    for (auto& kv : g_ctx) {
      // comms already aborted (and map cleared) when g_comm_failed
      for (auto& ck : kv.second->comms) ncclCommDestroy(ck.second);
      for (auto& ok : kv.second->oneshot) {
        for (void* p : ok.second.opened) (void)hipIpcCloseMemHandle(p);
        if (ok.second.my_staging) (void)hipFree(ok.second.my_staging);
        if (ok.second.my_flags) (void)hipFree(ok.second.my_flags);
      }
      for (auto& hk : kv.second->hier) {
        if (!g_comm_failed) {
          if (hk.second.local) ncclCommDestroy(hk.second.local);
          if (hk.second.cross) ncclCommDestroy(hk.second.cross);
        }
      }
      delete kv.second;
    }
    g_ctx.clear();
  }
  g_comm_failed = false;  // elastic re-init starts clean
  g_bootstrapped.clear();
  g_oneshot_bootstrapped.clear();
  g_hier_bootstrapped.clear();
  std::lock_guard<std::mutex> g(g_event_mu);
  for (auto ev : g_event_pool) (void)hipEventDestroy(ev);
  g_event_pool.clear();
}

}  // namespace gpu
}  // namespace hvd
