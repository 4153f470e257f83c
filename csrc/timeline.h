// Chrome-trace timeline (reference: horovod/common/timeline.{cc,h} N20).
//
// Writes about:tracing / Perfetto-compatible JSON.  Each tensor (by name) is
// modeled as a "process" row, as in the reference; op activities are emitted
// as B/E duration events from a dedicated writer thread so the hot path only
// pays a queue push.
#pragma once

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdio>
#include <deque>
#include <mutex>
#include <string>
#include <thread>
#include <unordered_map>

#include "message.h"

namespace hvd {

class Timeline {
 public:
  // rank names the output file (rank 0 writes `path`, others `path.<rank>`)
  Timeline(const std::string& path, int rank);
  ~Timeline();

  void OpStart(const Response& r);
  void OpEnd(const Response& r);
  // Named activity within an op (e.g. MEMCPY_IN_FUSION_BUFFER, RCCL_ALLREDUCE)
  void Activity(const std::string& tensor, const std::string& activity,
                int64_t start_us, int64_t end_us);
  void NegotiateStart(const std::string& tensor);
  void NegotiateEnd(const std::string& tensor);
  void Marker(const std::string& name);

  // Drain the writer, close the JSON array and the file.  SYNCHRONOUS and
  // idempotent: hvd.stop_timeline() calls this so the file is valid,
  // complete JSON the moment the call returns (late events from references
  // still held by other threads become no-ops).

  void Finalize();

  int64_t NowUs() const {
    return std::chrono::duration_cast<std::chrono::microseconds>(
               std::chrono::steady_clock::now() - t0_)
        .count();
  }

 private:
  struct Record {
    std::string json;
  };
  int PidOf(const std::string& tensor);
  void Push(std::string json);
  void WriterLoop();

  std::chrono::steady_clock::time_point t0_;
  FILE* file_ = nullptr;
  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<Record> queue_;
  // pids_ is touched from the background thread AND the GPU finalizer
  // thread (Activity) — needs its own lock
  std::mutex pid_mu_;
  std::unordered_map<std::string, int> pids_;
  int next_pid_ = 1;
  std::atomic<bool> stop_{false};
  std::atomic<bool> finalized_{false};
  bool first_ = true;
  std::thread writer_;
};

}  // namespace hvd
