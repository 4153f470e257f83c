// Online autotuner for {fusion threshold, cycle time}.
//
// Re-design of the reference ParameterManager (horovod/common/
// parameter_manager.{cc,h} + optim/bayesian_optimization.cc: Gaussian
// process + expected improvement over {fusion-threshold-MB, cycle-time-ms}
// scored by throughput).  Same idea, self-contained implementation: a small
// RBF-kernel GP on the normalized 2-D space, EI maximized over random
// candidates, warm-started from a coarse grid.  Runs at the coordinator;
// winning parameters are broadcast to every rank as TUNE responses so
// fast-path fusion stays bit-identical across ranks.
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
  };

  Autotuner(int64_t fusion_bytes, double cycle_time_ms, std::string log_path);

  // Feed the bytes processed this cycle; returns true when a new parameter
  // proposal is ready (window elapsed) — fetch it with current().
  bool Record(int64_t bytes, double now_sec);

  Params current() const { return current_; }
  bool done() const { return done_; }

 private:
  double Score(double bytes, double secs) const { return bytes / secs; }
  void Observe(double score);
  Params Propose();
  // GP machinery (normalized inputs in [0,1]^2)
  double Kernel(const double* a, const double* b) const;
  void PosteriorStats(const double* x, double& mu, double& sigma) const;

  Params current_;
  Params best_;
  double best_score_ = -1.0;
  bool done_ = false;

  std::vector<std::array<double, 2>> xs_;  // normalized samples
  std::vector<double> ys_;                 // scores (normalized later)

  // sampling window
  double window_start_ = -1.0;
  int64_t window_bytes_ = 0;
  double window_sec_ = 3.0;
  int max_samples_ = 24;
  size_t grid_idx_ = 0;
  std::vector<Params> grid_;
  std::mt19937 rng_{12345};
  std::string log_path_;
};

}  // namespace hvd
