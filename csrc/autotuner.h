// Online autotuner for {fusion threshold, cycle time, one-shot crossover,
// cache on/off}.
//
// Re-design of the reference ParameterManager (horovod/common/
// parameter_manager.{cc,h} + optim/bayesian_optimization.cc: Gaussian
// process + expected improvement over the continuous dimensions, with the
// categorical choices the reference tunes as booleans replaced by the
// xGMI-relevant ones: the ring-vs-one-shot crossover threshold (a log2
// continuous dimension, active when HOROVOD_ONESHOT_ALLREDUCE is on) and
// the response-cache enable bit (a {0,1} coordinate in the same RBF
// kernel — distance 1 in normalized space keeps the two arms nearly
// independent, the self-contained analogue of the reference's per-category
// GPs).  Runs at the coordinator; winning parameters are broadcast to
// every rank as TUNE responses so fast-path fusion stays bit-identical
// across ranks.  After convergence the tuner keeps watching the
// steady-state score and REOPENS tuning when the workload shifts
// (reference re-tunes on workload change; round-1 froze forever).
#pragma once

#include <array>
#include <cstdint>
#include <random>
#include <string>
#include <vector>

namespace hvd {

class Autotuner {
 public:
  struct Params {
    int64_t fusion_bytes;
    double cycle_time_ms;
    int64_t oneshot_threshold;  // bytes; ring-vs-one-shot crossover
    bool cache_enabled;
  };

  Autotuner(int64_t fusion_bytes, double cycle_time_ms,
            int64_t oneshot_threshold, std::string log_path);

  // Feed the bytes processed this cycle; returns true when a new parameter
  // proposal is ready (window elapsed) — fetch it with current().
  bool Record(int64_t bytes, double now_sec);
  // Post-convergence monitor: keeps scoring windows and reopens tuning when
  // the workload signature drifts.  Returns true when tuning reopened.
  bool Watch(int64_t bytes, double now_sec);

  Params current() const { return current_; }
  bool done() const { return done_; }

 private:
  double Score(double bytes, double secs) const { return bytes / secs; }
  void Observe(double score);
  Params Propose();
  // GP machinery (normalized inputs in [0,1]^2)
  double Kernel(const double* a, const double* b) const;
  void PosteriorStats(const double* x, double& mu, double& sigma) const;
  static constexpr int kDims = 4;  // fusion, cycle, oneshot-log2, cache

  Params current_;
  Params best_;
  double best_score_ = -1.0;
  bool done_ = false;

  std::vector<std::array<double, 4>> xs_;  // normalized samples
  std::vector<double> ys_;                 // scores (normalized later)

  // sampling window
  double window_start_ = -1.0;
  int64_t window_bytes_ = 0;
  double window_sec_ = 3.0;
  int max_samples_ = 24;
  bool tune_oneshot_ = false;   // HOROVOD_ONESHOT_ALLREDUCE set
  // post-convergence drift detection -> reopen tuning
  int drift_windows_ = 0;
  size_t grid_idx_ = 0;
  std::vector<Params> grid_;
  std::mt19937 rng_{12345};
  std::string log_path_;
};

}  // namespace hvd
