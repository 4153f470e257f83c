#include "core.h"

#include <pthread.h>
#include <sched.h>

#include <chrono>
#include <cstdio>
#include <cstring>
#include <stdexcept>

#include "gpu.h"
#include "logging.h"
#include "timeline.h"

namespace hvd {

namespace {
std::string SetKey(int32_t set_id, const std::string& name) {
  return std::to_string(set_id) + ":" + name;
}
}  // namespace

// ---------------------------------------------------------------------------
// TensorQueue
// ---------------------------------------------------------------------------
Status TensorQueue::Add(Request req, TensorTableEntry entry) {
  std::lock_guard<std::mutex> g(mu_);
  std::string key = SetKey(req.process_set_id, req.name);
  if (table_.count(key)) {
    return Status::InvalidArgument(
        "Duplicate tensor name in queue: " + req.name +
        ". This happens if multiple outstanding collectives share a name; pass "
        "a unique name= to each call.");
  }
  table_.emplace(std::move(key), std::move(entry));
  messages_.push_back(std::move(req));
  return Status::OK();
}

Status TensorQueue::AddMulti(std::vector<Request>& reqs,
                             std::vector<TensorTableEntry>& entries) {
  std::lock_guard<std::mutex> g(mu_);
  for (auto& r : reqs) {
    if (table_.count(SetKey(r.process_set_id, r.name)))
      return Status::InvalidArgument("Duplicate tensor name in queue: " + r.name);
  }
  for (size_t i = 0; i < reqs.size(); ++i) {
    table_.emplace(SetKey(reqs[i].process_set_id, reqs[i].name),
                   std::move(entries[i]));
    messages_.push_back(std::move(reqs[i]));
  }
  return Status::OK();
}

bool TensorQueue::has_messages() const {
  std::lock_guard<std::mutex> g(mu_);
  return !messages_.empty();
}

std::vector<Request> TensorQueue::PopMessages() {
  std::lock_guard<std::mutex> g(mu_);
  std::vector<Request> out(messages_.begin(), messages_.end());
  messages_.clear();
  return out;
}

bool TensorQueue::PopEntry(int32_t set_id, const std::string& name,
                           TensorTableEntry& out) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = table_.find(SetKey(set_id, name));
  if (it == table_.end()) return false;
  out = std::move(it->second);
  table_.erase(it);
  return true;
}

void TensorQueue::FailAll(const Status& s) {
  std::lock_guard<std::mutex> g(mu_);
  for (auto& kv : table_) {
    if (kv.second.callback) kv.second.callback(s, kv.second);
  }
  table_.clear();
  messages_.clear();
}

size_t TensorQueue::size() const {
  std::lock_guard<std::mutex> g(mu_);
  return table_.size();
}

// ---------------------------------------------------------------------------
// HandleManager
// ---------------------------------------------------------------------------
int HandleManager::Allocate(int n_outputs) {
  std::lock_guard<std::mutex> g(mu_);
  int h = next_++;
  auto hs = std::make_shared<HandleState>();
  hs->outputs.resize(n_outputs);
  handles_[h] = hs;
  return h;
}

std::shared_ptr<HandleState> HandleManager::Get(int handle) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = handles_.find(handle);
  return it == handles_.end() ? nullptr : it->second;
}

bool HandleManager::Poll(int handle) {
  auto h = Get(handle);
  if (!h) return true;
  std::lock_guard<std::mutex> g(h->mu);
  return h->done;
}

Status HandleManager::Wait(int handle, std::vector<at::Tensor>& outputs,
                           at::Tensor& extra, int32_t* result_int) {
  auto h = Get(handle);
  if (!h) return Status::InvalidArgument("unknown handle");
  std::unique_lock<std::mutex> lk(h->mu);
  h->cv.wait(lk, [&] { return h->done; });
  outputs = h->outputs;
  extra = h->extra;
  if (result_int) *result_int = h->result_int;
  Status s = h->status;
  lk.unlock();
  Drop(handle);
  return s;
}

void HandleManager::MarkDone(int handle, const Status& s) {
  auto h = Get(handle);
  if (!h) return;
  std::lock_guard<std::mutex> g(h->mu);
  h->status = s;
  h->done = true;
  h->cv.notify_all();
}

void HandleManager::FailAll(const Status& s) {
  std::lock_guard<std::mutex> g(mu_);
  for (auto& kv : handles_) {
    std::lock_guard<std::mutex> g2(kv.second->mu);
    if (!kv.second->done) {
      kv.second->status = s;
      kv.second->done = true;
      kv.second->cv.notify_all();
    }
  }
}

void HandleManager::Drop(int handle) {
  std::lock_guard<std::mutex> g(mu_);
  handles_.erase(handle);
}

// ---------------------------------------------------------------------------
// Global state
// ---------------------------------------------------------------------------
GlobalState& State() {
  static GlobalState st;
  return st;
}

std::shared_ptr<Timeline> GetTimeline(GlobalState& st) {
  return std::atomic_load(&st.timeline);
}
void SetTimeline(GlobalState& st, std::shared_ptr<Timeline> t) {
  std::atomic_store(&st.timeline, std::move(t));
}

bool IsInitialized() { return State().initialized; }

void FlushCycle() {
  auto& st = State();
  {
    std::lock_guard<std::mutex> g(st.pace_mu);
    st.flush_requested = true;
  }
  st.pace_cv.notify_one();
}

namespace {

void FailEntries(std::vector<TensorTableEntry>& entries, const Status& s) {
  for (auto& e : entries)
    if (e.callback) e.callback(s, e);
}

std::vector<std::vector<int64_t>> ParseShapes(const Response& r) {
  std::vector<std::vector<int64_t>> shapes;
  const int64_t* p = r.tensor_shapes.data();
  const int64_t* end = p + r.tensor_shapes.size();
  while (p < end) {
    int64_t nd = *p++;
    shapes.emplace_back(p, p + nd);
    p += nd;
  }
  return shapes;
}

at::Tensor FlatPrescaled(const TensorTableEntry& e, DataType wire) {
  at::Tensor t = e.tensor.flatten();
  auto wire_t = DataTypeToTorch(wire);
  if (e.prescale != 1.0) {
    t = t.to(at::kDouble).mul_(e.prescale).to(wire_t);
  } else if (t.scalar_type() != wire_t) {
    t = t.to(wire_t);
  } else if (!t.is_contiguous()) {
    t = t.contiguous();
  }
  return t;
}

void UnpackInto(TensorTableEntry& e, at::Tensor flat_slice) {
  at::Tensor src = flat_slice;
  if (e.postscale != 1.0) src = src.to(at::kDouble).mul_(e.postscale);
  if (!e.output.defined()) e.output = at::empty_like(e.tensor);
  // copy_ handles layout + dtype (output may be strided)
  e.output.copy_(src.reshape(e.output.sizes()));
}

// Adasum pairwise combine: a = a*(1 - dot/(2|a|^2)) + b*(1 - dot/(2|b|^2)),
// per tensor within the fused buffer (reference: adasum.h:347-412
// FusedPairwiseReduceWithComm — per-tensor coefficient isolation).
void AdasumCombineInPlace(at::Tensor a, at::Tensor b,
                          const std::vector<int64_t>& offsets,
                          const std::vector<int64_t>& counts) {
  for (size_t i = 0; i < offsets.size(); ++i) {
    auto av = a.narrow(0, offsets[i], counts[i]);
    auto bv = b.narrow(0, offsets[i], counts[i]);
    double dot = av.dot(bv).item<double>();
    double na = av.dot(av).item<double>();
    double nb = bv.dot(bv).item<double>();
    double acoef = na > 0 ? 1.0 - dot / (2.0 * na) : 1.0;
    double bcoef = nb > 0 ? 1.0 - dot / (2.0 * nb) : 1.0;
    av.mul_(acoef).add_(bv, bcoef);
  }
}

// --------------------------- CPU data plane --------------------------------
// All CPU collectives are star-routed through global rank 0 and are GLOBAL
// lock-step frames: every rank (member or not) participates in exactly one
// Gather and one Bcast (plus ScatterFrames for alltoall), so control-plane
// framing can never interleave with data.  This is the correctness tier —
// the performance tier is RCCL over xGMI (gpu.cc).

void PerTensorLayout(const Response& resp, std::vector<int64_t>& offsets,
                     std::vector<int64_t>& counts) {
  auto shapes = ParseShapes(resp);
  int64_t off = 0;
  for (auto& sh : shapes) {
    int64_t n = 1;
    for (auto d : sh) n *= d;
    offsets.push_back(off);
    counts.push_back(n);
    off += n;
  }
}

// ---- mesh algorithms ------------------------------------------------------
// Direct member-to-member data plane (MeshComm): ring allreduce, pairwise-
// exchange allgather/alltoall/reducescatter, binomial broadcast.  Each rank
// moves ~2(n-1)/n x data for allreduce instead of 2n x through rank 0.
// Reference analogue: mpi_operations.cc MPI_Allreduce/Allgatherv/Alltoallv.
// Only set MEMBERS participate; frames ride per-pair sockets in response
// order, so no interleaving with the control star is possible.

bool UseMesh(GlobalState& st, const ProcessSetInfo& set) {
  return st.mesh.alive() && st.size > 1 && set.ranks.size() > 1;
}

void ApplyReduce(at::Tensor acc, at::Tensor incoming, ReduceOp op) {
  switch (op) {
    case ReduceOp::MIN: at::minimum_out(acc, acc, incoming); break;
    case ReduceOp::MAX: at::maximum_out(acc, acc, incoming); break;
    case ReduceOp::PRODUCT: acc.mul_(incoming); break;
    default: acc.add_(incoming); break;
  }
}

// In-place ring allreduce on a flat contiguous tensor (wire dtype).
// Phase 1: ring reduce-scatter (n-1 steps); phase 2: ring allgather.
// Every element's accumulation order is a pure function of its segment, so
// results are bit-identical on every rank.
void MeshRingAllreduce(GlobalState& st, const ProcessSetInfo& set,
                       at::Tensor buf, ReduceOp op) {
  int n = (int)set.ranks.size();
  int li = set.local_index(st.rank);
  int64_t numel = buf.numel();
  int64_t esz = buf.element_size();
  if (numel == 0 || n <= 1) return;
  auto seg_off = [&](int s) {
    int64_t base = numel / n, rem = numel % n;
    return (int64_t)s * base + std::min<int64_t>(s, rem);
  };
  auto seg_len = [&](int s) {
    return numel / n + (s < numel % n ? 1 : 0);
  };
  int right = set.ranks[(li + 1) % n];
  int left = set.ranks[(li - 1 + n) % n];
  char* base = (char*)buf.data_ptr();
  at::Tensor scratch = at::empty({numel / n + 1}, buf.options());
  // reduce-scatter: after step k, segment (li-k-1) holds partials of k+2 ranks
  for (int step = 0; step < n - 1; ++step) {
    int ss = (li - step + n) % n;         // segment I send
    int rs = (li - step - 1 + n) % n;     // segment I receive+reduce
    int64_t slen = seg_len(ss), rlen = seg_len(rs);
    st.mesh.SendRecv2(right, base + seg_off(ss) * esz, (size_t)(slen * esz),
                      left, scratch.data_ptr(), (size_t)(rlen * esz));
    if (rlen)
      ApplyReduce(buf.narrow(0, seg_off(rs), rlen),
                  scratch.narrow(0, 0, rlen), op);
  }
  // allgather: circulate the finished segments
  for (int step = 0; step < n - 1; ++step) {
    int ss = (li + 1 - step + n) % n;     // finished segment I hold
    int rs = (li - step + n) % n;         // segment I receive
    int64_t slen = seg_len(ss), rlen = seg_len(rs);
    st.mesh.SendRecv2(right, base + seg_off(ss) * esz, (size_t)(slen * esz),
                      left, base + seg_off(rs) * esz, (size_t)(rlen * esz));
  }
}

// Distributed VHDD Adasum over the mesh (CPU twin of gpu.cc's
// ExecuteAdasumRSVHDD; reference: adasum_mpi.cc point-to-point VHDD).
// Works entirely within set MEMBERS (scalar group sums ride the mesh, not
// the star), so subset Adasum is supported.  All math in float64, same
// combine tree as the AdasumCombineInPlace golden.
void MeshAdasum(GlobalState& st, const ProcessSetInfo& set,
                const Response& resp,
                std::vector<TensorTableEntry>& entries) {
  const int n = (int)set.ranks.size();
  const int li = set.local_index(st.rank);
  DataType wire = resp.dtype;
  std::vector<int64_t> offsets, counts;
  PerTensorLayout(resp, offsets, counts);
  const int64_t T = (int64_t)counts.size();

  std::vector<at::Tensor> flats;
  for (auto& e : entries) flats.push_back(FlatPrescaled(e, wire));
  at::Tensor work =
      (flats.size() == 1 ? flats[0] : at::cat(flats)).to(at::kDouble);
  if (!work.is_contiguous()) work = work.contiguous();
  const int64_t L = work.numel();
  double* wp = work.data_ptr<double>();

  int p = 1;
  while (p * 2 <= n) p *= 2;
  int levels = 0;
  while ((1 << levels) < p) ++levels;

  // fold the non-power-of-2 remainder: (i-p, i) pairs, full-vector combine
  if (li >= p) {
    st.mesh.Send(set.ranks[li - p], wp, (size_t)(L * 8));
  } else if (li + p < n) {
    at::Tensor peer = at::empty({L}, work.options());
    st.mesh.Recv(set.ranks[li + p], peer.data_ptr(), (size_t)(L * 8));
    AdasumCombineInPlace(work, peer, offsets, counts);
  }

  // per-owner final piece ranges are a pure function of the split path
  auto final_range = [&](int r) {
    int64_t S = 0, len = L;
    for (int k = 0; k < levels; ++k) {
      int64_t half = len / 2;
      if ((r >> k) & 1) {
        S += half;
        len -= half;
      } else {
        len = half;
      }
    }
    return std::pair<int64_t, int64_t>(S, len);
  };

  int64_t S = 0, len = L;
  if (li < p) {
    at::Tensor scratch = at::empty({(L + 1) / 2}, work.options());
    for (int k = 0; k < levels; ++k) {
      int stride = 1 << k;
      int partner = li ^ stride;
      bool lower = li < partner;
      int64_t half = len / 2;
      int64_t my_len = lower ? half : len - half;
      int64_t myS = lower ? S : S + half;
      int64_t th_len = len - my_len;
      int64_t thS = lower ? S + half : S;
      st.mesh.SendRecv2(set.ranks[partner], wp + thS, (size_t)(th_len * 8),
                        set.ranks[partner], scratch.data_ptr(),
                        (size_t)(my_len * 8));
      double* sp = scratch.data_ptr<double>();
      // canonical per-tensor partial dots over my piece (A = lower's data)
      std::vector<double> part((size_t)(3 * T), 0.0);
      for (int64_t t = 0; t < T; ++t) {
        int64_t lo = std::max(myS, offsets[t]);
        int64_t hi = std::min(myS + my_len, offsets[t] + counts[t]);
        for (int64_t i = lo; i < hi; ++i) {
          double mine = wp[i], theirs = sp[i - myS];
          double a = lower ? mine : theirs, b = lower ? theirs : mine;
          part[t * 3 + 0] += a * b;
          part[t * 3 + 1] += a * a;
          part[t * 3 + 2] += b * b;
        }
      }
      // group-sum the scalars over the 2*stride ranks sharing this vector
      // (pairwise mesh exchange; sum in ascending member order so every
      // rank derives bit-identical coefficients)
      int group = 2 * stride;
      int g0 = (li / group) * group;
      std::vector<std::vector<double>> by_member((size_t)group);
      by_member[li - g0] = part;
      std::vector<double> inbox((size_t)(3 * T));
      for (int j = 1; j < group; ++j) {
        // XOR pairing: symmetric (x^j partners with x), so the same-peer
        // full-duplex exchange is matched on both sides (group is a
        // power of 2)
        int peer_li = g0 + (((li - g0) ^ j) & (group - 1));
        st.mesh.SendRecv2(set.ranks[peer_li], part.data(),
                          part.size() * 8, set.ranks[peer_li], inbox.data(),
                          inbox.size() * 8);
        by_member[peer_li - g0] = inbox;
      }
      std::vector<double> dots((size_t)(3 * T), 0.0);
      for (int m = 0; m < group; ++m)
        for (size_t i = 0; i < dots.size(); ++i)
          dots[i] += by_member[m].empty() ? 0.0 : by_member[m][i];
      // combine my piece: result = acoef*A + bcoef*B
      for (int64_t t = 0; t < T; ++t) {
        double dot = dots[t * 3], na = dots[t * 3 + 1], nb = dots[t * 3 + 2];
        double ac = na > 0 ? 1.0 - dot / (2.0 * na) : 1.0;
        double bc = nb > 0 ? 1.0 - dot / (2.0 * nb) : 1.0;
        double cm = lower ? ac : bc;  // coefficient of MY data
        double cr = lower ? bc : ac;  // coefficient of the received data
        int64_t lo = std::max(myS, offsets[t]);
        int64_t hi = std::min(myS + my_len, offsets[t] + counts[t]);
        for (int64_t i = lo; i < hi; ++i)
          wp[i] = cm * wp[i] + cr * sp[i - myS];
      }
      S = myS;
      len = my_len;
    }
  }

  // regather: each of the p owners ships its final piece to every other
  // member (folded ranks included), pair transfers in fixed (r, dst) order
  for (int r = 0; r < p; ++r) {
    auto [Sr, lr] = final_range(r);
    if (lr == 0) continue;
    if (li == r) {
      for (int dst = 0; dst < n; ++dst)
        if (dst != li)
          st.mesh.Send(set.ranks[dst], wp + Sr, (size_t)(lr * 8));
    } else {
      st.mesh.Recv(set.ranks[r], wp + Sr, (size_t)(lr * 8));
    }
  }

  at::Tensor out = work.to(DataTypeToTorch(wire));
  for (size_t i = 0; i < entries.size(); ++i)
    UnpackInto(entries[i], out.narrow(0, offsets[i], counts[i]));
}

void MeshAllreduce(GlobalState& st, const ProcessSetInfo& set,
                   const Response& resp,
                   std::vector<TensorTableEntry>& entries) {
  DataType wire = resp.dtype;
  std::vector<int64_t> offsets, counts;
  PerTensorLayout(resp, offsets, counts);
  std::vector<at::Tensor> flats;
  for (auto& e : entries) flats.push_back(FlatPrescaled(e, wire));
  at::Tensor buf;
  bool in_place = false;
  if (flats.size() == 1) {
    auto& e = entries[0];
    // the ring reduces IN PLACE: only alias the user's storage when the op
    // is explicitly in-place (output == tensor) with no dtype conversion;
    // otherwise clone, or a second allreduce of the same input would read
    // already-reduced values
    in_place = e.output.defined() &&
               e.output.data_ptr() == e.tensor.data_ptr() &&
               flats[0].data_ptr() == e.tensor.data_ptr() &&
               e.postscale == 1.0;
    buf = in_place ? flats[0] : flats[0].clone();
  } else {
    buf = at::cat(flats);  // always a copy
  }
  MeshRingAllreduce(st, set, buf, resp.reduce_op);
  if (!in_place)
    for (size_t i = 0; i < entries.size(); ++i)
      UnpackInto(entries[i], buf.narrow(0, offsets[i], counts[i]));
}

void MeshAllgather(GlobalState& st, const ProcessSetInfo& set,
                   const Response& resp,
                   std::vector<TensorTableEntry>& entries) {
  int n = (int)set.ranks.size();
  int li = set.local_index(st.rank);
  auto& e = entries[0];
  at::Tensor in = e.tensor.contiguous();
  auto shapes = ParseShapes(resp);
  int64_t row_elems = 1;
  for (size_t d = 1; d < shapes[0].size(); ++d) row_elems *= shapes[0][d];
  int64_t esz = in.element_size();
  int64_t total0 = 0;
  std::vector<int64_t> offs(n);
  for (int r = 0; r < n; ++r) {
    offs[r] = total0;
    total0 += resp.tensor_sizes[r];
  }
  std::vector<int64_t> out_shape = shapes[0];
  if (out_shape.empty()) out_shape = {total0};
  else out_shape[0] = total0;
  e.output = at::empty(out_shape, e.tensor.options());
  char* out_base = (char*)e.output.data_ptr();
  // my own contribution in place
  if (resp.tensor_sizes[li])
    std::memcpy(out_base + offs[li] * row_elems * esz, in.data_ptr(),
                (size_t)(resp.tensor_sizes[li] * row_elems * esz));
  // pairwise exchange: step s sends my rows to li+s, receives li-s's rows
  for (int s = 1; s < n; ++s) {
    int to = (li + s) % n, from = (li - s + n) % n;
    st.mesh.SendRecv2(
        set.ranks[to], in.data_ptr(),
        (size_t)(resp.tensor_sizes[li] * row_elems * esz), set.ranks[from],
        out_base + offs[from] * row_elems * esz,
        (size_t)(resp.tensor_sizes[from] * row_elems * esz));
  }
  int nn = (int)resp.tensor_sizes.size();
  e.received_splits = at::empty({nn}, at::kLong);
  auto* gs = e.received_splits.data_ptr<int64_t>();
  for (int r = 0; r < nn; ++r) gs[r] = resp.tensor_sizes[r];
}

void MeshBroadcast(GlobalState& st, const ProcessSetInfo& set,
                   const Response& resp,
                   std::vector<TensorTableEntry>& entries) {
  int n = (int)set.ranks.size();
  int li = set.local_index(st.rank);
  int root_li = set.local_index(resp.root_rank);
  if (root_li < 0) root_li = 0;
  auto& e = entries[0];
  at::Tensor in = e.tensor.is_non_overlapping_and_dense()
                      ? e.tensor
                      : e.tensor.contiguous();
  bool is_root = li == root_li;
  at::Tensor stage;
  if (is_root) {
    stage = in;
  } else {
    stage = (e.output.defined() && e.output.is_contiguous() &&
             e.output.numel() == in.numel())
                ? e.output
                : at::empty({in.numel()}, in.options());
  }
  size_t bytes = (size_t)(in.numel() * in.element_size());
  // binomial tree on virtual ranks (root at 0)
  int v = (li - root_li + n) % n;
  int mask = 1;
  while (mask < n && (v & (mask - 1)) == 0) {
    if (v & mask) {
      int src = (v - mask + root_li) % n;
      st.mesh.Recv(set.ranks[src], stage.data_ptr(), bytes);
      break;
    }
    mask <<= 1;
  }
  while ((mask >>= 1) > 0) {
    if (v + mask < n) {
      int dst = (v + mask + root_li) % n;
      st.mesh.Send(set.ranks[dst], stage.data_ptr(), bytes);
    }
  }
  if (!e.output.defined()) {
    e.output = is_root ? in : stage.reshape(in.sizes());
  } else if (e.output.data_ptr() != stage.data_ptr()) {
    e.output.copy_(is_root ? in : stage.reshape(e.output.sizes()));
  } else if (!is_root && !e.output.sizes().equals(in.sizes()) &&
             e.output.numel() == in.numel()) {
    e.output = e.output.reshape(in.sizes());
  }
}

void MeshAlltoall(GlobalState& st, const ProcessSetInfo& set,
                  const Response& resp,
                  std::vector<TensorTableEntry>& entries) {
  int n = (int)set.ranks.size();
  int li = set.local_index(st.rank);
  auto& e = entries[0];
  at::Tensor in = e.tensor.contiguous();
  int64_t row_elems = 1;
  for (int d = 1; d < e.tensor.dim(); ++d) row_elems *= e.tensor.size(d);
  int64_t esz = in.element_size();
  auto srows = [&](int i, int j) {  // rows sender i ships to receiver j
    return resp.tensor_sizes[(size_t)i * n + j];
  };
  std::vector<int64_t> send_off(n), recv_off(n);
  int64_t so = 0, ro = 0;
  for (int r = 0; r < n; ++r) {
    send_off[r] = so;
    so += srows(li, r);
    recv_off[r] = ro;
    ro += srows(r, li);
  }
  std::vector<int64_t> out_shape(e.tensor.sizes().begin(),
                                 e.tensor.sizes().end());
  if (out_shape.empty()) out_shape = {ro};
  else out_shape[0] = ro;
  e.output = at::empty(out_shape, e.tensor.options());
  char* in_base = (char*)in.data_ptr();
  char* out_base = (char*)e.output.data_ptr();
  // self block
  if (srows(li, li))
    std::memcpy(out_base + recv_off[li] * row_elems * esz,
                in_base + send_off[li] * row_elems * esz,
                (size_t)(srows(li, li) * row_elems * esz));
  for (int s = 1; s < n; ++s) {
    int to = (li + s) % n, from = (li - s + n) % n;
    st.mesh.SendRecv2(set.ranks[to],
                      in_base + send_off[to] * row_elems * esz,
                      (size_t)(srows(li, to) * row_elems * esz),
                      set.ranks[from],
                      out_base + recv_off[from] * row_elems * esz,
                      (size_t)(srows(from, li) * row_elems * esz));
  }
  e.received_splits = at::empty({n}, at::kLong);
  auto* rs = e.received_splits.data_ptr<int64_t>();
  for (int r = 0; r < n; ++r) rs[r] = srows(r, li);
}

void MeshReducescatter(GlobalState& st, const ProcessSetInfo& set,
                       const Response& resp,
                       std::vector<TensorTableEntry>& entries) {
  int n = (int)set.ranks.size();
  int li = set.local_index(st.rank);
  auto& e = entries[0];
  DataType wire = resp.dtype;
  at::Tensor flat = FlatPrescaled(e, wire).contiguous();
  auto shapes = ParseShapes(resp);
  std::vector<int64_t> shape = shapes[0];
  int64_t first = shape.empty() ? 1 : shape[0];
  int64_t row_elems = 1;
  for (size_t d = 1; d < shape.size(); ++d) row_elems *= shape[d];
  int64_t esz = flat.element_size();
  int64_t base_rows = first / n, rem = first % n;
  auto rows_of = [&](int r) { return base_rows + (r < rem ? 1 : 0); };
  auto roff = [&](int r) {
    return (int64_t)r * base_rows + std::min<int64_t>(r, rem);
  };
  // my accumulator = my own slice (accumulation order: own + peers in
  // (li-s) order — identical on every rank for a given element? No: each
  // element is reduced by exactly one rank, so per-element order is fixed.)
  at::Tensor acc =
      flat.narrow(0, roff(li) * row_elems, rows_of(li) * row_elems).clone();
  at::Tensor scratch = at::empty_like(acc);
  char* in_base = (char*)flat.data_ptr();
  for (int s = 1; s < n; ++s) {
    int to = (li + s) % n, from = (li - s + n) % n;
    st.mesh.SendRecv2(set.ranks[to], in_base + roff(to) * row_elems * esz,
                      (size_t)(rows_of(to) * row_elems * esz),
                      set.ranks[from], scratch.data_ptr(),
                      (size_t)(rows_of(li) * row_elems * esz));
    if (acc.numel()) ApplyReduce(acc, scratch, resp.reduce_op);
  }
  std::vector<int64_t> out_shape = shape;
  if (out_shape.empty()) out_shape = {rows_of(li)};
  else out_shape[0] = rows_of(li);
  e.output = at::empty(out_shape, e.tensor.options());
  at::Tensor src = acc;
  if (e.postscale != 1.0) src = src.to(at::kDouble).mul_(e.postscale);
  e.output.flatten().copy_(src.to(e.output.scalar_type()));
}

void CPUAllreduce(GlobalState& st, const Response& resp,
                  std::vector<TensorTableEntry>& entries, bool member) {
  auto& set = st.controller->process_set(resp.process_set_id);
  if (UseMesh(st, set)) {
    // direct links; non-members do nothing (no star frames)
    if (member && !entries.empty()) {
      if (resp.type == ResponseType::ADASUM)
        MeshAdasum(st, set, resp, entries);
      else
        MeshAllreduce(st, set, resp, entries);
    }
    return;
  }
  DataType wire = resp.dtype;
  auto wire_t = DataTypeToTorch(wire);
  std::string payload;
  std::vector<int64_t> offsets, counts;
  PerTensorLayout(resp, offsets, counts);
  if (member && !entries.empty()) {
    std::vector<at::Tensor> flats;
    for (auto& e : entries) flats.push_back(FlatPrescaled(e, wire));
    at::Tensor buf = flats.size() == 1 ? flats[0] : at::cat(flats);
    payload.assign((const char*)buf.data_ptr(), buf.numel() * buf.element_size());
  }
  auto gathered = st.comm.Gather(payload);
  std::string result;
  if (st.comm.is_root()) {
    std::vector<at::Tensor> bufs;
    for (int r : set.ranks) {
      if (r < (int)gathered.size() && !gathered[r].empty()) {
        auto& s = gathered[r];
        int64_t n = (int64_t)(s.size() / DataTypeSize(wire));
        bufs.push_back(at::from_blob((void*)s.data(), {n}, wire_t).clone());
      }
    }
    if (!bufs.empty()) {
      at::Tensor acc;
      if (resp.type == ResponseType::ADASUM) {
        std::vector<at::Tensor> work;
        for (auto& b : bufs) work.push_back(b.to(at::kDouble));
        size_t p = 1;
        while (p * 2 <= work.size()) p *= 2;
        for (size_t i = p; i < work.size(); ++i)
          AdasumCombineInPlace(work[i - p], work[i], offsets, counts);
        work.resize(p);
        for (size_t stride = 1; stride < p; stride *= 2)
          for (size_t i = 0; i + stride < p; i += 2 * stride)
            AdasumCombineInPlace(work[i], work[i + stride], offsets, counts);
        acc = work[0].to(wire_t);
      } else {
        acc = bufs[0];
        for (size_t i = 1; i < bufs.size(); ++i) {
          switch (resp.reduce_op) {
            case ReduceOp::MIN: acc = at::minimum(acc, bufs[i]); break;
            case ReduceOp::MAX: acc = at::maximum(acc, bufs[i]); break;
            case ReduceOp::PRODUCT: acc = acc.mul(bufs[i]); break;
            default: acc = acc.add(bufs[i]); break;
          }
        }
      }
      result.assign((const char*)acc.data_ptr(), acc.numel() * acc.element_size());
    }
  }
  result = st.comm.Bcast(result);
  if (member && !entries.empty()) {
    int64_t n = (int64_t)(result.size() / DataTypeSize(wire));
    at::Tensor buf = at::from_blob((void*)result.data(), {n}, wire_t);
    for (size_t i = 0; i < entries.size(); ++i)
      UnpackInto(entries[i], buf.narrow(0, offsets[i], counts[i]));
  }
}

void CPUAllgather(GlobalState& st, const Response& resp,
                  std::vector<TensorTableEntry>& entries, bool member) {
  auto& set = st.controller->process_set(resp.process_set_id);
  if (UseMesh(st, set)) {
    if (member && !entries.empty()) MeshAllgather(st, set, resp, entries);
    return;
  }
  std::string payload;
  if (member && !entries.empty()) {
    at::Tensor t = entries[0].tensor.contiguous();
    payload.assign((const char*)t.data_ptr(), t.numel() * t.element_size());
  }
  auto gathered = st.comm.Gather(payload);
  std::string result;
  if (st.comm.is_root()) {
    for (int r : set.ranks)
      if (r < (int)gathered.size()) result += gathered[r];
  }
  result = st.comm.Bcast(result);
  if (member && !entries.empty()) {
    auto& e = entries[0];
    auto shapes = ParseShapes(resp);
    std::vector<int64_t> out_shape = shapes[0];
    int64_t total0 = 0;
    for (auto s : resp.tensor_sizes) total0 += s;
    if (out_shape.empty()) out_shape = {total0};
    else out_shape[0] = total0;
    e.output = at::empty(out_shape, e.tensor.options());
    std::memcpy(e.output.data_ptr(), result.data(),
                std::min((size_t)(e.output.numel() * e.output.element_size()),
                         result.size()));
    // per-rank first-dim sizes (see gpu.cc allgather note)
    int n = (int)resp.tensor_sizes.size();
    e.received_splits = at::empty({n}, at::kLong);
    auto* gs = e.received_splits.data_ptr<int64_t>();
    for (int r = 0; r < n; ++r) gs[r] = resp.tensor_sizes[r];
  }
}

void CPUBroadcast(GlobalState& st, const Response& resp,
                  std::vector<TensorTableEntry>& entries, bool member) {
  {
    auto& set = st.controller->process_set(resp.process_set_id);
    if (UseMesh(st, set)) {
      if (member && !entries.empty()) MeshBroadcast(st, set, resp, entries);
      return;
    }
  }
  std::string payload;
  if (member && !entries.empty() && resp.root_rank == st.rank) {
    at::Tensor t = entries[0].tensor.contiguous();
    payload.assign((const char*)t.data_ptr(), t.numel() * t.element_size());
  }
  auto gathered = st.comm.Gather(payload);
  std::string result;
  if (st.comm.is_root() && resp.root_rank < (int)st.size &&
      resp.root_rank < (int)gathered.size())
    result = gathered[resp.root_rank];
  result = st.comm.Bcast(result);
  if (member && !entries.empty()) {
    auto& e = entries[0];
    if (!e.output.defined()) e.output = at::empty_like(e.tensor);
    if ((size_t)(e.output.numel() * e.output.element_size()) == result.size()) {
      if (e.output.is_contiguous()) {
        std::memcpy(e.output.data_ptr(), result.data(), result.size());
      } else {
        int64_t n = e.output.numel();
        at::Tensor src = at::from_blob((void*)result.data(), {n},
                                       e.output.options().device(at::kCPU));
        e.output.copy_(src.reshape(e.output.sizes()));
      }
    }
  }
}

void CPUAlltoall(GlobalState& st, const Response& resp,
                 std::vector<TensorTableEntry>& entries, bool member) {
  auto& set = st.controller->process_set(resp.process_set_id);
  if (UseMesh(st, set)) {
    if (member && !entries.empty()) MeshAlltoall(st, set, resp, entries);
    return;
  }
  int n = (int)set.ranks.size();
  std::string payload;
  if (member && !entries.empty()) {
    at::Tensor t = entries[0].tensor.contiguous();
    payload.assign((const char*)t.data_ptr(), t.numel() * t.element_size());
  }
  auto gathered = st.comm.Gather(payload);
  std::vector<std::string> outgoing((size_t)st.size);
  if (st.comm.is_root()) {
    auto shapes = ParseShapes(resp);
    int64_t row_bytes = (int64_t)DataTypeSize(resp.dtype);
    if (!shapes.empty())
      for (size_t d = 1; d < shapes[0].size(); ++d) row_bytes *= shapes[0][d];
    for (int i = 0; i < n; ++i) {
      int src_global = set.ranks[i];
      if (src_global >= (int)gathered.size()) continue;
      const std::string& buf = gathered[src_global];
      int64_t off = 0;
      for (int j = 0; j < n; ++j) {
        int64_t rows = resp.tensor_sizes[(size_t)i * n + j];
        int dst_global = set.ranks[j];
        outgoing[dst_global].append(buf.data() + off * row_bytes,
                                    (size_t)(rows * row_bytes));
        off += rows;
      }
    }
  }
  std::string mine = st.comm.ScatterFrames(outgoing);
  if (member && !entries.empty()) {
    auto& e = entries[0];
    int li = set.local_index(st.rank);
    int64_t total_rows = 0;
    for (int i = 0; i < n; ++i) total_rows += resp.tensor_sizes[(size_t)i * n + li];
    std::vector<int64_t> out_shape(e.tensor.sizes().begin(), e.tensor.sizes().end());
    if (out_shape.empty()) out_shape = {total_rows};
    else out_shape[0] = total_rows;
    e.output = at::empty(out_shape, e.tensor.options());
    std::memcpy(e.output.data_ptr(), mine.data(),
                std::min((size_t)(e.output.numel() * e.output.element_size()),
                         mine.size()));
    e.received_splits = at::empty({n}, at::kLong);
    auto* rs = e.received_splits.data_ptr<int64_t>();
    for (int i = 0; i < n; ++i) rs[i] = resp.tensor_sizes[(size_t)i * n + li];
  }
}

void CPUReducescatter(GlobalState& st, const Response& resp,
                      std::vector<TensorTableEntry>& entries, bool member) {
  auto& set = st.controller->process_set(resp.process_set_id);
  if (UseMesh(st, set)) {
    if (member && !entries.empty())
      MeshReducescatter(st, set, resp, entries);
    return;
  }
  int n = (int)set.ranks.size();
  DataType wire = resp.dtype;
  auto wire_t = DataTypeToTorch(wire);
  std::string payload;
  if (member && !entries.empty()) {
    auto f = FlatPrescaled(entries[0], wire);
    payload.assign((const char*)f.data_ptr(), f.numel() * f.element_size());
  }
  auto gathered = st.comm.Gather(payload);
  std::string result;
  if (st.comm.is_root()) {
    at::Tensor acc;
    for (int r : set.ranks) {
      if (r >= (int)gathered.size() || gathered[r].empty()) continue;
      int64_t cnt = (int64_t)(gathered[r].size() / DataTypeSize(wire));
      auto t = at::from_blob((void*)gathered[r].data(), {cnt}, wire_t).clone();
      acc = acc.defined() ? acc.add(t) : t;
    }
    if (acc.defined())
      result.assign((const char*)acc.data_ptr(), acc.numel() * acc.element_size());
  }
  result = st.comm.Bcast(result);
  if (member && !entries.empty()) {
    auto& e = entries[0];
    auto shapes = ParseShapes(resp);
    std::vector<int64_t> shape = shapes[0];
    int64_t first = shape.empty() ? 1 : shape[0];
    int64_t base = first / n, rem = first % n;
    int li = set.local_index(st.rank);
    int64_t my_rows = base + (li < rem ? 1 : 0);
    int64_t row_off = li * base + std::min<int64_t>(li, rem);
    int64_t row_elems = 1;
    for (size_t d = 1; d < shape.size(); ++d) row_elems *= shape[d];
    std::vector<int64_t> out_shape = shape;
    if (out_shape.empty()) out_shape = {my_rows};
    else out_shape[0] = my_rows;
    int64_t cnt = (int64_t)(result.size() / DataTypeSize(wire));
    at::Tensor full = at::from_blob((void*)result.data(), {cnt}, wire_t);
    e.output = at::empty(out_shape, e.tensor.options());
    at::Tensor src = full.narrow(0, row_off * row_elems, my_rows * row_elems);
    if (e.postscale != 1.0) src = src.to(at::kDouble).mul_(e.postscale);
    e.output.flatten().copy_(src.to(e.output.scalar_type()));
  }
}

// ------------------------- dispatch ----------------------------------------

void PerformOperation(GlobalState& st, Response& resp) {
  if (!st.controller->has_process_set(resp.process_set_id)) return;
  auto& set = st.controller->process_set(resp.process_set_id);
  bool member = set.contains(st.rank);
  bool joined;
  {
    std::lock_guard<std::mutex> g(st.join_mu);
    joined = st.local_joined.count(resp.process_set_id) &&
             st.local_joined[resp.process_set_id];
  }

  if (resp.type == ResponseType::TUNE) {
    if (resp.tensor_sizes.size() >= 2) {
      bool cache_on = resp.tensor_sizes.size() < 4 || resp.tensor_sizes[3];
      st.controller->ApplyTune(resp.tensor_sizes[0],
                               resp.tensor_sizes[1] / 1000.0, cache_on);
      if (resp.tensor_sizes.size() >= 3)
        gpu::SetOneshotThreshold(resp.tensor_sizes[2]);
    }
    return;
  }
  if (resp.type == ResponseType::JOIN) {
    TensorTableEntry e;
    if (st.queue.PopEntry(resp.process_set_id, "join", e)) {
      e.join_result = resp.last_joined_rank;
      if (e.callback) e.callback(Status::OK(), e);
    }
    std::lock_guard<std::mutex> g(st.join_mu);
    st.local_joined[resp.process_set_id] = false;
    return;
  }
  if (resp.type == ResponseType::BARRIER) {
    TensorTableEntry e;
    if (st.queue.PopEntry(resp.process_set_id, "barrier", e) && e.callback)
      e.callback(Status::OK(), e);
    return;
  }

  std::vector<TensorTableEntry> entries;
  if (member && !joined) {
    for (auto& name : resp.names) {
      TensorTableEntry e;
      if (st.queue.PopEntry(resp.process_set_id, name, e)) {
        entries.push_back(std::move(e));
      } else if (resp.type != ResponseType::BARRIER) {
        std::fprintf(stderr, "[horovod_amd] missing entry for %s\n", name.c_str());
      }
    }
  } else if (member && joined &&
             (resp.type == ResponseType::ALLREDUCE ||
              resp.type == ResponseType::ADASUM ||
              resp.type == ResponseType::ALLGATHER ||
              resp.type == ResponseType::BROADCAST ||
              resp.type == ResponseType::REDUCESCATTER ||
              resp.type == ResponseType::ALLTOALL)) {
    // Zero substitution for a joined rank (reference: tensor_queue.cc
    // GetTensorEntriesFromResponse:125-141 substitutes for EVERY op type):
    // allreduce/adasum/reducescatter contribute zeros, allgather/alltoall
    // contribute zero rows (the alltoall split-matrix row for a joined rank
    // is already all zeros, but it must still post its recvs), broadcast
    // receives into a scratch buffer — all so the rank still participates
    // in the RCCL collective its comm peers will issue.
    auto shapes = ParseShapes(resp);
    int dev;
    {
      std::lock_guard<std::mutex> g(st.join_mu);
      dev = st.join_device.count(resp.process_set_id)
                ? st.join_device[resp.process_set_id]
                : CPU_DEVICE_ID;
    }
    int set_n = (int)set.ranks.size();
    int my_li = set.local_index(st.rank);
    for (size_t i = 0; i < resp.names.size(); ++i) {
      TensorTableEntry e;
      e.name = resp.names[i];
      auto opts = at::TensorOptions()
                      .dtype(DataTypeToTorch(resp.dtype))
                      .device(dev == CPU_DEVICE_ID ? at::Device(at::kCPU)
                                                   : at::Device(at::kCUDA, dev));
      std::vector<int64_t> shape = shapes[i];
      // first-dim substitution must match what the RESPONSE says this rank
      // contributes (reference: AllocateZeros(response.tensor_sizes()[i]),
      // tensor_queue.cc:133): a CACHED response negotiated before the join
      // still carries the old sizes, and peers size their windows by them —
      // a hardcoded 0 here segfaults the data plane on stale-size rows
      if (resp.type == ResponseType::ALLGATHER && !shape.empty()) {
        int64_t rows = 0;
        size_t per = resp.names.size()
                         ? resp.tensor_sizes.size() / resp.names.size()
                         : 0;
        if (my_li >= 0 && per && (size_t)my_li < per)
          rows = resp.tensor_sizes[i * per + my_li];
        shape[0] = rows;
      } else if (resp.type == ResponseType::ALLTOALL && !shape.empty()) {
        int64_t rows = 0;
        if (my_li >= 0 &&
            resp.tensor_sizes.size() >= (size_t)(set_n * set_n))
          for (int j = 0; j < set_n; ++j)
            rows += resp.tensor_sizes[(size_t)my_li * set_n + j];
        shape[0] = rows;
      }
      e.tensor = at::zeros(shape, opts);
      e.device = dev;
      e.process_set_id = resp.process_set_id;
      entries.push_back(std::move(e));
    }
  }

  if (resp.type == ResponseType::ERROR) {
    FailEntries(entries, Status::PreconditionError(resp.error_msg));
    return;
  }

  auto tl = GetTimeline(st);
  if (tl) {
    for (auto& n : resp.names) tl->NegotiateEnd(n);
    tl->OpStart(resp);
  }

  // Response device is the normalized marker (-1 CPU, -2 GPU) — the GPU path
  // must run on every rank (even relay-only non-members) so the RCCL comm
  // bootstrap's star frames stay lock-step.
  bool gpu_op = resp.device != CPU_DEVICE_ID;
  try {
    if (gpu_op) {
      gpu::Execute(st, resp, entries);  // async; completion via finalizer
      // close the submit span; the finalizer's Activity records the actual
      // device-side completion window
      if (tl) tl->OpEnd(resp);
    } else {
      switch (resp.type) {
        case ResponseType::ALLREDUCE:
        case ResponseType::ADASUM:
          CPUAllreduce(st, resp, entries, member);
          break;
        case ResponseType::ALLGATHER:
          CPUAllgather(st, resp, entries, member);
          break;
        case ResponseType::BROADCAST:
          CPUBroadcast(st, resp, entries, member);
          break;
        case ResponseType::ALLTOALL:
          CPUAlltoall(st, resp, entries, member);
          break;
        case ResponseType::REDUCESCATTER:
          CPUReducescatter(st, resp, entries, member);
          break;
        default:
          break;
      }
      FailEntries(entries, Status::OK());
      if (tl) tl->OpEnd(resp);
    }
  } catch (const std::exception& ex) {
    // a RCCL call failing because comms were aborted is the elastic-
    // recoverable signal, not an internal bug
    FailEntries(entries, gpu::CommsFailed() ? Status::Aborted(ex.what())
                                            : Status::UnknownError(ex.what()));
    if (tl) tl->OpEnd(resp);
  }
}

void Abort(GlobalState& st, const std::string& why) {
  st.aborted = true;
  st.abort_reason = why;
  HVD_LOG(ERROR, "background loop aborted: %s", why.c_str());
  // Tear down the GPU data plane FIRST: a peer that died mid-collective
  // leaves RCCL kernels hung on the comm stream; ncclCommAbort unblocks
  // them so the finalizer can fail their handles (and so WaitAllPending in
  // the shutdown path terminates).
  gpu::AbortComms(why);
  auto s = Status::Aborted(why);
  st.queue.FailAll(s);
  st.handles.FailAll(s);
}

void SetThreadAffinity() {
  // reference: parse_and_set_affinity (common.cc) / HOROVOD_THREAD_AFFINITY
  const char* aff = std::getenv("HOROVOD_THREAD_AFFINITY");
  if (!aff) return;
  int cpu = atoi(aff) + State().local_rank;
  cpu_set_t set;
  CPU_ZERO(&set);
  CPU_SET(cpu % CPU_SETSIZE, &set);
  pthread_setaffinity_np(pthread_self(), sizeof(set), &set);
}

void BackgroundLoop(GlobalState& st) {
  SetThreadAffinity();
  const bool mark_cycles = std::getenv("HOROVOD_TIMELINE_MARK_CYCLES") != nullptr;
  uint64_t cycle_no = 0;
  // Shutdown protocol: a graceful exit needs every rank's agreement (AND bit
  // in the cache-sync round) so no peer is left mid-collective; but a rank
  // must never hang forever on a dead peer — after the grace window it
  // leaves unilaterally and survivors see a socket error (-> their pending
  // ops fail with HorovodInternalError, the elastic recovery signal).
  const char* grace_env = std::getenv("HOROVOD_SHUTDOWN_GRACE_SECONDS");
  double grace = grace_env ? atof(grace_env) : 5.0;
  std::chrono::steady_clock::time_point shutdown_since{};
  while (true) {
    auto cycle_start = std::chrono::steady_clock::now();
    if (st.shutdown_requested) {
      if (shutdown_since.time_since_epoch().count() == 0)
        shutdown_since = cycle_start;
      else if (std::chrono::duration<double>(cycle_start - shutdown_since)
                   .count() > grace)
        break;
    }
    auto reqs = st.queue.PopMessages();
    ResponseList rl;
    try {
      rl = st.controller->RunCycle(std::move(reqs), st.shutdown_requested);
    } catch (const std::exception& ex) {
      Abort(st, ex.what());
      break;
    }
    for (auto& resp : rl.responses) PerformOperation(st, resp);
    ++cycle_no;
    if (mark_cycles && !rl.responses.empty()) {
      auto tl = GetTimeline(st);
      if (tl) tl->Marker("CYCLE_" + std::to_string(cycle_no));
    }
    if (rl.shutdown) break;
    // pace the cycle cadence: gradient bursts accumulate for up to
    // cycle_time so they fuse into large buckets; a flush (synchronize())
    // cuts the wait for the step's trailing bucket.
    {
      auto elapsed = std::chrono::duration<double, std::milli>(
                         std::chrono::steady_clock::now() - cycle_start)
                         .count();
      double remain = st.controller->cycle_time_ms() - elapsed;
      if (remain > 0) {
        std::unique_lock<std::mutex> lk(st.pace_mu);
        st.pace_cv.wait_for(lk,
                            std::chrono::duration<double, std::milli>(remain),
                            [&] { return st.flush_requested; });
        st.flush_requested = false;
      }
    }
  }
  gpu::WaitAllPending();
  st.shutting_down = true;
}

}  // namespace

// ---------------------------------------------------------------------------
// Lifecycle
// ---------------------------------------------------------------------------
void InitHorovod(int rank, int size, int local_rank, int local_size,
                 int cross_rank, int cross_size, const std::string& addr,
                 int port, const ControllerConfig& cfg) {
  auto& st = State();
  if (st.initialized) return;
  st.rank = rank;
  st.size = size;
  st.local_rank = local_rank;
  st.local_size = local_size;
  st.cross_rank = cross_rank;
  st.cross_size = cross_size;
  st.aborted = false;
  st.shutdown_requested = false;
  st.shutting_down = false;
  st.abort_reason.clear();
  double start_timeout = 120.0;
  if (const char* e = std::getenv("HOROVOD_START_TIMEOUT"))
    start_timeout = atof(e);
  st.comm.Init(rank, size, addr, port, start_timeout);
  // CPU data plane: full mesh of member-to-member links (ring allreduce &
  // friends).  HOROVOD_CPU_STAR=1 keeps the round-1 star fallback.
  if (!std::getenv("HOROVOD_CPU_STAR"))
    st.mesh.Init(st.comm, addr, start_timeout);
  st.controller.reset(new Controller(&st.comm, rank, size, cfg));
  if (cfg.timeline_enabled) {
    const char* tf = std::getenv("HOROVOD_TIMELINE");
    if (tf) SetTimeline(st, std::make_shared<Timeline>(tf, rank));
  }
  st.bg_thread = std::thread([&st] { BackgroundLoop(st); });
  st.initialized = true;
  HVD_LOG(INFO, "initialized: rank %d/%d (local %d/%d), controller %s:%d",
          rank, size, local_rank, local_size, addr.c_str(), port);
}

void InterruptHorovod(const std::string& why) {
  auto& st = State();
  if (!st.initialized || st.aborted) return;
  // prompt elastic scale-down: a displaced worker (or a host-side watchdog)
  // fails all in-flight and future collectives with HorovodInternalError so
  // the retry loop resets without waiting for the next commit()
  Abort(st, why);
}

void ShutdownHorovod() {
  auto& st = State();
  if (!st.initialized) return;
  st.shutdown_requested = true;
  if (st.bg_thread.joinable()) st.bg_thread.join();
  gpu::Shutdown();
  st.mesh.Shutdown();
  st.comm.Shutdown();
  st.controller.reset();
  SetTimeline(st, nullptr);
  st.local_joined.clear();
  st.join_device.clear();
  st.initialized = false;
}

// ---------------------------------------------------------------------------
// Enqueue API
// ---------------------------------------------------------------------------
namespace {

Request MakeRequest(RequestType type, const std::string& name,
                    const at::Tensor& t, ReduceOp op, double pre, double post,
                    int root, int32_t set_id, DataType wire_dtype) {
  auto& st = State();
  Request r;
  r.type = type;
  r.rank = st.rank;
  r.name = name;
  r.dtype = wire_dtype;
  r.shape.assign(t.sizes().begin(), t.sizes().end());
  r.root_rank = root;
  r.reduce_op = op;
  r.prescale = pre;
  r.postscale = post;
  r.process_set_id = set_id;
  // Normalized device marker: -1 CPU, -2 GPU (executor uses the entry's real
  // local device index; see controller.h design notes).
  r.device = t.is_cpu() ? CPU_DEVICE_ID : -2;
  return r;
}

TensorTableEntry MakeEntry(const std::string& name, at::Tensor t, at::Tensor out,
                           ReduceOp op, double pre, double post, int root,
                           int32_t set_id) {
  TensorTableEntry e;
  e.name = name;
  e.tensor = t;
  e.output = out;
  e.device = t.is_cpu() ? CPU_DEVICE_ID : (int)t.get_device();
  e.root_rank = root;
  e.reduce_op = op;
  e.prescale = pre;
  e.postscale = post;
  e.process_set_id = set_id;
  // Default: NO per-tensor ready event.  hipEventRecord onto the busy
  // compute stream costs ~120 us apiece on ROCm (161 gradients = +20 ms per
  // ResNet-50 step, measured); instead the executor records ONE event per
  // fused response on the default stream, which covers every already-
  // enqueued producer.  Users driving collectives from non-default streams
  // can restore per-tensor events with HOROVOD_PER_TENSOR_READY_EVENTS=1.
  static const bool per_tensor_events =
      std::getenv("HOROVOD_PER_TENSOR_READY_EVENTS") != nullptr;
  if (e.device != CPU_DEVICE_ID && per_tensor_events)
    e.ready_event = gpu::RecordReadyEvent(e.device);
  return e;
}

int EnqueueImpl(std::vector<Request> reqs, std::vector<TensorTableEntry> entries) {
  auto& st = State();
  if (!st.initialized)
    throw std::runtime_error(
        "horovod_amd has not been initialized; call hvd.init() first.");
  if (st.aborted)
    throw std::runtime_error("HorovodInternalError: " + st.abort_reason);
  int handle = st.handles.Allocate((int)entries.size());
  auto hs = st.handles.Get(handle);
  auto remaining = std::make_shared<std::atomic<int>>((int)entries.size());
  for (size_t i = 0; i < entries.size(); ++i) {
    entries[i].callback = [hs, remaining, i](const Status& s, TensorTableEntry& e) {
      {
        std::lock_guard<std::mutex> g(hs->mu);
        if (hs->outputs.size() <= i) hs->outputs.resize(i + 1);
        hs->outputs[i] = e.output.defined() ? e.output : e.tensor;
        if (e.received_splits.defined()) hs->extra = e.received_splits;
        if (e.join_result >= 0) hs->result_int = e.join_result;
        if (!s.ok() && hs->status.ok()) hs->status = s;
      }
      if (remaining->fetch_sub(1) == 1) {
        std::lock_guard<std::mutex> g(hs->mu);
        hs->done = true;
        hs->cv.notify_all();
      }
    };
  }
  Status added = st.queue.AddMulti(reqs, entries);
  if (!added.ok()) st.handles.MarkDone(handle, added);
  return handle;
}

}  // namespace

int EnqueueAllreduceMulti(std::vector<at::Tensor> tensors,
                          std::vector<at::Tensor> outputs,
                          std::vector<std::string> names, ReduceOp op,
                          double prescale, double postscale,
                          int32_t process_set_id, DataType wire_dtype) {
  std::vector<Request> reqs;
  std::vector<TensorTableEntry> entries;
  for (size_t i = 0; i < tensors.size(); ++i) {
    auto req = MakeRequest(RequestType::ALLREDUCE, names[i], tensors[i], op,
                           prescale, postscale, -1, process_set_id, wire_dtype);
    if (tensors.size() > 1) {
      // strict group identity (reference: GroupTable) — the whole group
      // negotiates and executes as one fused unit
      req.group_key = names[0];
      req.group_size = (int32_t)tensors.size();
    }
    reqs.push_back(std::move(req));
    entries.push_back(MakeEntry(names[i], tensors[i], outputs[i], op, prescale,
                                postscale, -1, process_set_id));
  }
  return EnqueueImpl(std::move(reqs), std::move(entries));
}

int EnqueueAllgather(at::Tensor tensor, const std::string& name,
                     int32_t process_set_id) {
  auto req = MakeRequest(RequestType::ALLGATHER, name, tensor, ReduceOp::SUM, 1.0,
                         1.0, -1, process_set_id, DataTypeFromTorch(tensor.scalar_type()));
  auto e = MakeEntry(name, tensor, at::Tensor(), ReduceOp::SUM, 1.0, 1.0, -1,
                     process_set_id);
  std::vector<Request> reqs{std::move(req)};
  std::vector<TensorTableEntry> entries{std::move(e)};
  return EnqueueImpl(std::move(reqs), std::move(entries));
}

int EnqueueBroadcast(at::Tensor tensor, at::Tensor output, int root_rank,
                     const std::string& name, int32_t process_set_id) {
  auto req = MakeRequest(RequestType::BROADCAST, name, tensor, ReduceOp::SUM, 1.0,
                         1.0, root_rank, process_set_id,
                         DataTypeFromTorch(tensor.scalar_type()));
  auto e = MakeEntry(name, tensor, output, ReduceOp::SUM, 1.0, 1.0, root_rank,
                     process_set_id);
  std::vector<Request> reqs{std::move(req)};
  std::vector<TensorTableEntry> entries{std::move(e)};
  return EnqueueImpl(std::move(reqs), std::move(entries));
}

int EnqueueAlltoall(at::Tensor tensor, at::Tensor splits, const std::string& name,
                    int32_t process_set_id) {
  auto req = MakeRequest(RequestType::ALLTOALL, name, tensor, ReduceOp::SUM, 1.0,
                         1.0, -1, process_set_id,
                         DataTypeFromTorch(tensor.scalar_type()));
  auto e = MakeEntry(name, tensor, at::Tensor(), ReduceOp::SUM, 1.0, 1.0, -1,
                     process_set_id);
  if (splits.defined() && splits.numel() > 0) {
    auto s = splits.to(at::kLong).contiguous();
    auto* p = s.data_ptr<int64_t>();
    req.splits.assign(p, p + s.numel());
    e.splits = req.splits;
  }
  std::vector<Request> reqs{std::move(req)};
  std::vector<TensorTableEntry> entries{std::move(e)};
  return EnqueueImpl(std::move(reqs), std::move(entries));
}

int EnqueueReducescatter(at::Tensor tensor, const std::string& name, ReduceOp op,
                         double prescale, double postscale,
                         int32_t process_set_id) {
  auto req = MakeRequest(RequestType::REDUCESCATTER, name, tensor, op, prescale,
                         postscale, -1, process_set_id,
                         DataTypeFromTorch(tensor.scalar_type()));
  auto e = MakeEntry(name, tensor, at::Tensor(), op, prescale, postscale, -1,
                     process_set_id);
  std::vector<Request> reqs{std::move(req)};
  std::vector<TensorTableEntry> entries{std::move(e)};
  return EnqueueImpl(std::move(reqs), std::move(entries));
}

int EnqueueJoin(int device, int32_t process_set_id) {
  auto& st = State();
  {
    std::lock_guard<std::mutex> g(st.join_mu);
    st.local_joined[process_set_id] = true;
    st.join_device[process_set_id] = device;
  }
  Request r;
  r.type = RequestType::JOIN;
  r.rank = st.rank;
  r.name = "join";
  r.process_set_id = process_set_id;
  r.device = device == CPU_DEVICE_ID ? CPU_DEVICE_ID : -2;
  TensorTableEntry e;
  e.name = "join";
  e.device = device;
  e.process_set_id = process_set_id;
  std::vector<Request> reqs{std::move(r)};
  std::vector<TensorTableEntry> entries{std::move(e)};
  return EnqueueImpl(std::move(reqs), std::move(entries));
}

int EnqueueBarrier(int32_t process_set_id) {
  auto& st = State();
  Request r;
  r.type = RequestType::BARRIER;
  r.rank = st.rank;
  r.name = "barrier";
  r.process_set_id = process_set_id;
  TensorTableEntry e;
  e.name = "barrier";
  e.process_set_id = process_set_id;
  std::vector<Request> reqs{std::move(r)};
  std::vector<TensorTableEntry> entries{std::move(e)};
  return EnqueueImpl(std::move(reqs), std::move(entries));
}

}  // namespace hvd
