#include "timeline.h"

namespace hvd {

namespace {
const char* TypeName(ResponseType t) {
  switch (t) {
    case ResponseType::ALLREDUCE: return "ALLREDUCE";
    case ResponseType::ALLGATHER: return "ALLGATHER";
    case ResponseType::BROADCAST: return "BROADCAST";
    case ResponseType::JOIN: return "JOIN";
    case ResponseType::ADASUM: return "ADASUM";
    case ResponseType::ALLTOALL: return "ALLTOALL";
    case ResponseType::BARRIER: return "BARRIER";
    case ResponseType::REDUCESCATTER: return "REDUCESCATTER";
    case ResponseType::ERROR: return "ERROR";
    case ResponseType::TUNE: return "TUNE";
  }
  return "?";
}

std::string Escape(const std::string& s) {
  std::string out;
  for (char c : s) {
    if (c == '"' || c == '\\') out.push_back('\\');
    out.push_back(c);
  }
  return out;
}
}  // namespace

Timeline::Timeline(const std::string& path, int rank)
    : t0_(std::chrono::steady_clock::now()) {
  std::string p = path;
  if (rank != 0) p += "." + std::to_string(rank);
  file_ = std::fopen(p.c_str(), "w");
  if (file_) std::fputs("[\n", file_);
  writer_ = std::thread([this] { WriterLoop(); });
}

Timeline::~Timeline() { Finalize(); }

void Timeline::Finalize() {
  bool expected = false;
  if (!finalized_.compare_exchange_strong(expected, true)) return;
  stop_ = true;
  cv_.notify_all();
  if (writer_.joinable()) writer_.join();
  std::lock_guard<std::mutex> g(mu_);
  if (file_) {
    // close the JSON array so the file is valid the moment stop returns
    // (reference keeps validity by seek-back patching after every write)
    std::fputs("\n]\n", file_);
    std::fclose(file_);
    file_ = nullptr;
  }
}

int Timeline::PidOf(const std::string& tensor) {
  std::unique_lock<std::mutex> lk(pid_mu_);
  auto it = pids_.find(tensor);
  if (it != pids_.end()) return it->second;
  int pid = next_pid_++;
  pids_[tensor] = pid;
  lk.unlock();
  char buf[512];
  std::snprintf(buf, sizeof(buf),
                "{\"ph\":\"M\",\"name\":\"process_name\",\"pid\":%d,\"args\":{"
                "\"name\":\"%s\"}}",
                pid, Escape(tensor).c_str());
  Push(buf);
  std::snprintf(buf, sizeof(buf),
                "{\"ph\":\"M\",\"name\":\"process_sort_index\",\"pid\":%d,"
                "\"args\":{\"sort_index\":%d}}",
                pid, pid);
  Push(buf);
  return pid;
}

void Timeline::Push(std::string json) {
  if (finalized_) return;  // late event from a still-held reference
  std::lock_guard<std::mutex> g(mu_);
  queue_.push_back({std::move(json)});
  cv_.notify_one();
}

void Timeline::WriterLoop() {
  while (true) {
    std::deque<Record> batch;
    {
      std::unique_lock<std::mutex> lk(mu_);
      cv_.wait(lk, [&] { return stop_ || !queue_.empty(); });
      batch.swap(queue_);
      if (batch.empty() && stop_) break;
    }
    if (!file_) continue;
    for (auto& r : batch) {
      if (!first_) std::fputs(",\n", file_);
      first_ = false;
      std::fputs(r.json.c_str(), file_);
    }
    std::fflush(file_);
  }
}

void Timeline::OpStart(const Response& r) {
  int64_t ts = NowUs();
  for (auto& name : r.names) {
    int pid = PidOf(name);
    char buf[512];
    std::snprintf(buf, sizeof(buf),
                  "{\"ph\":\"B\",\"name\":\"%s\",\"pid\":%d,\"tid\":0,\"ts\":%lld}",
                  TypeName(r.type), pid, (long long)ts);
    Push(buf);
  }
}

void Timeline::OpEnd(const Response& r) {
  int64_t ts = NowUs();
  for (auto& name : r.names) {
    int pid = PidOf(name);
    char buf[256];
    std::snprintf(buf, sizeof(buf),
                  "{\"ph\":\"E\",\"pid\":%d,\"tid\":0,\"ts\":%lld}", pid,
                  (long long)ts);
    Push(buf);
  }
}

void Timeline::Activity(const std::string& tensor, const std::string& activity,
                        int64_t start_us, int64_t end_us) {
  int pid = PidOf(tensor);
  char buf[512];
  std::snprintf(buf, sizeof(buf),
                "{\"ph\":\"X\",\"name\":\"%s\",\"pid\":%d,\"tid\":1,\"ts\":%lld,"
                "\"dur\":%lld}",
                Escape(activity).c_str(), pid, (long long)start_us,
                (long long)(end_us - start_us));
  Push(buf);
}

void Timeline::NegotiateStart(const std::string& tensor) {
  int pid = PidOf(tensor);
  char buf[256];
  std::snprintf(buf, sizeof(buf),
                "{\"ph\":\"B\",\"name\":\"NEGOTIATE\",\"pid\":%d,\"tid\":0,\"ts\":"
                "%lld}",
                pid, (long long)NowUs());
  Push(buf);
}

void Timeline::NegotiateEnd(const std::string& tensor) {
  int pid = PidOf(tensor);
  char buf[256];
  std::snprintf(buf, sizeof(buf),
                "{\"ph\":\"E\",\"pid\":%d,\"tid\":0,\"ts\":%lld}", pid,
                (long long)NowUs());
  Push(buf);
}

void Timeline::Marker(const std::string& name) {
  char buf[512];
  std::snprintf(buf, sizeof(buf),
                "{\"ph\":\"i\",\"name\":\"%s\",\"pid\":0,\"tid\":0,\"ts\":%lld,"
                "\"s\":\"g\"}",
                Escape(name).c_str(), (long long)NowUs());
  Push(buf);
}

}  // namespace hvd
