#include "autotuner.h"

#include <algorithm>
#include <array>
#include <cmath>
#include <cstdio>

namespace hvd {

namespace {
// normalization bounds: fusion in [1 MB, 256 MB] (log2), cycle in
// [0.1 ms, 20 ms] (log10), one-shot crossover in [64 KB, 8 MB] (log2)
constexpr double kFusLo = 0.0, kFusHi = 8.0;       // log2(MB)
constexpr double kCycLo = -1.0, kCycHi = 1.30103;  // log10(ms)
constexpr double kOsLo = 16.0, kOsHi = 23.0;       // log2(bytes)

std::array<double, 4> Normalize(const Autotuner::Params& p) {
  double f = std::log2((double)p.fusion_bytes / (1 << 20));
  double c = std::log10(p.cycle_time_ms);
  double o = std::log2((double)std::max<int64_t>(p.oneshot_threshold, 1));
  return {(f - kFusLo) / (kFusHi - kFusLo), (c - kCycLo) / (kCycHi - kCycLo),
          (o - kOsLo) / (kOsHi - kOsLo), p.cache_enabled ? 1.0 : 0.0};
}

Autotuner::Params Denormalize(const std::array<double, 4>& x) {
  double f = kFusLo + x[0] * (kFusHi - kFusLo);
  double c = kCycLo + x[1] * (kCycHi - kCycLo);
  double o = kOsLo + x[2] * (kOsHi - kOsLo);
  Autotuner::Params p;
  p.fusion_bytes = (int64_t)(std::pow(2.0, f) * (1 << 20));
  p.cycle_time_ms = std::pow(10.0, c);
  p.oneshot_threshold = (int64_t)std::pow(2.0, o);
  p.cache_enabled = x[3] >= 0.5;
  return p;
}
}  // namespace

Autotuner::Autotuner(int64_t fusion_bytes, double cycle_time_ms,
                     int64_t oneshot_threshold, std::string log_path)
    : log_path_(std::move(log_path)) {
  tune_oneshot_ = std::getenv("HOROVOD_ONESHOT_ALLREDUCE") != nullptr;
  current_ = {fusion_bytes, cycle_time_ms, oneshot_threshold, true};
  best_ = current_;
  if (const char* w = std::getenv("HOROVOD_AUTOTUNE_WINDOW_SECONDS"))
    window_sec_ = atof(w);
  // coarse warm-start grid (reference seeds with a grid before the GP takes
  // over, parameter_manager.cc:44-61); the cache-off and one-shot arms get
  // a few dedicated probes so their region of the GP is observed
  for (double mb : {8.0, 32.0, 64.0, 128.0}) {
    for (double ms : {0.2, 1.0, 5.0}) {
      grid_.push_back({(int64_t)(mb * (1 << 20)), ms, oneshot_threshold,
                       true});
    }
  }
  grid_.push_back({64ll << 20, 1.0, oneshot_threshold, false});
  if (tune_oneshot_) {
    grid_.push_back({64ll << 20, 1.0, 256ll << 10, true});
    grid_.push_back({64ll << 20, 1.0, 8ll << 20, true});
  }
}

bool Autotuner::Record(int64_t bytes, double now_sec) {
  if (done_) return false;
  if (window_start_ < 0) {
    window_start_ = now_sec;
    window_bytes_ = 0;
  }
  window_bytes_ += bytes;
  double span = now_sec - window_start_;
  if (span < window_sec_ || window_bytes_ == 0) return false;
  Observe(Score((double)window_bytes_, span));
  window_start_ = now_sec;
  window_bytes_ = 0;
  if ((int)ys_.size() >= max_samples_) {
    current_ = best_;
    done_ = true;
    if (!log_path_.empty()) {
      FILE* f = std::fopen(log_path_.c_str(), "a");
      if (f) {
        std::fprintf(f, "CONVERGED,%lld,%f,%lld,%d,%f\n",
                     (long long)best_.fusion_bytes, best_.cycle_time_ms,
                     (long long)best_.oneshot_threshold,
                     (int)best_.cache_enabled, best_score_);
        std::fclose(f);
      }
    }
    return true;
  }
  current_ = Propose();
  return true;
}

bool Autotuner::Watch(int64_t bytes, double now_sec) {
  // post-convergence drift detection (reference: ParameterManager re-tunes
  // when the workload changes; round-1 stayed done_ forever).  Three
  // consecutive windows >50% away from the converged score reopen tuning.
  if (window_start_ < 0) {
    window_start_ = now_sec;
    window_bytes_ = 0;
  }
  window_bytes_ += bytes;
  double span = now_sec - window_start_;
  if (span < window_sec_ || window_bytes_ == 0) return false;
  double score = Score((double)window_bytes_, span);
  window_start_ = now_sec;
  window_bytes_ = 0;
  double rel = std::abs(score - best_score_) / std::max(best_score_, 1e-9);
  drift_windows_ = rel > 0.5 ? drift_windows_ + 1 : 0;
  if (drift_windows_ >= 3) {
    drift_windows_ = 0;
    done_ = false;
    xs_.clear();
    ys_.clear();
    best_score_ = -1.0;
    grid_idx_ = 0;
    if (!log_path_.empty()) {
      FILE* f = std::fopen(log_path_.c_str(), "a");
      if (f) {
        std::fprintf(f, "REOPENED,%f\n", score);
        std::fclose(f);
      }
    }
    return true;  // publish current params; tuning resumes next windows
  }
  return false;
}

void Autotuner::Observe(double score) {
  xs_.push_back(Normalize(current_));
  ys_.push_back(score);
  if (score > best_score_) {
    best_score_ = score;
    best_ = current_;
  }
  if (!log_path_.empty()) {
    FILE* f = std::fopen(log_path_.c_str(), "a");
    if (f) {
      std::fprintf(f, "%lld,%f,%lld,%d,%f\n",
                   (long long)current_.fusion_bytes, current_.cycle_time_ms,
                   (long long)current_.oneshot_threshold,
                   (int)current_.cache_enabled, score);
      std::fclose(f);
    }
  }
}

double Autotuner::Kernel(const double* a, const double* b) const {
  const double ls = 0.2;  // RBF length scale in normalized space
  double d2 = 0;
  for (int i = 0; i < kDims; ++i) d2 += (a[i] - b[i]) * (a[i] - b[i]);
  return std::exp(-d2 / (2 * ls * ls));
}

void Autotuner::PosteriorStats(const double* x, double& mu,
                               double& sigma) const {
  // GP posterior with noise 1e-3 on scores normalized to [0,1].
  size_t n = xs_.size();
  if (n == 0) {
    mu = 0;
    sigma = 1;
    return;
  }
  double ymax = *std::max_element(ys_.begin(), ys_.end());
  double ymin = *std::min_element(ys_.begin(), ys_.end());
  double span = std::max(ymax - ymin, 1e-9);

  // build K + sI and solve K a = y via Cholesky (n <= 24: cheap)
  std::vector<double> K(n * n);
  for (size_t i = 0; i < n; ++i)
    for (size_t j = 0; j < n; ++j)
      K[i * n + j] = Kernel(xs_[i].data(), xs_[j].data()) +
                     (i == j ? 1e-3 : 0.0);
  std::vector<double> L(n * n, 0.0);
  for (size_t i = 0; i < n; ++i) {
    for (size_t j = 0; j <= i; ++j) {
      double s = K[i * n + j];
      for (size_t k = 0; k < j; ++k) s -= L[i * n + k] * L[j * n + k];
      if (i == j)
        L[i * n + i] = std::sqrt(std::max(s, 1e-12));
      else
        L[i * n + j] = s / L[j * n + j];
    }
  }
  auto solve = [&](std::vector<double> b) {
    for (size_t i = 0; i < n; ++i) {
      for (size_t k = 0; k < i; ++k) b[i] -= L[i * n + k] * b[k];
      b[i] /= L[i * n + i];
    }
    for (size_t ii = n; ii-- > 0;) {
      for (size_t k = ii + 1; k < n; ++k) b[ii] -= L[k * n + ii] * b[k];
      b[ii] /= L[ii * n + ii];
    }
    return b;
  };
  std::vector<double> y(n);
  for (size_t i = 0; i < n; ++i) y[i] = (ys_[i] - ymin) / span;
  auto alpha = solve(y);

  std::vector<double> kx(n);
  for (size_t i = 0; i < n; ++i) kx[i] = Kernel(x, xs_[i].data());
  mu = 0;
  for (size_t i = 0; i < n; ++i) mu += kx[i] * alpha[i];
  auto v = solve(kx);
  double kxx = 1.0;
  double var = kxx;
  for (size_t i = 0; i < n; ++i) var -= kx[i] * v[i];
  sigma = std::sqrt(std::max(var, 1e-12));
}

Autotuner::Params Autotuner::Propose() {
  if (grid_idx_ < grid_.size()) {
    return grid_[grid_idx_++];
  }
  // expected improvement over random candidates
  double ymax = *std::max_element(ys_.begin(), ys_.end());
  double ymin = *std::min_element(ys_.begin(), ys_.end());
  double span = std::max(ymax - ymin, 1e-9);
  double fbest = (best_score_ - ymin) / span;

  std::uniform_real_distribution<double> u(0.0, 1.0);
  double best_ei = -1;
  std::array<double, 4> best_x{0.5, 0.5, 0.5, 1.0};
  for (int c = 0; c < 256; ++c) {
    // categorical cache bit sampled from {0,1}; one-shot dim pinned to the
    // current value unless that path is enabled
    std::array<double, 4> x{u(rng_), u(rng_),
                            tune_oneshot_ ? u(rng_)
                                          : Normalize(current_)[2],
                            u(rng_) < 0.5 ? 0.0 : 1.0};
    double mu, sigma;
    PosteriorStats(x.data(), mu, sigma);
    double z = (mu - fbest - 0.01) / sigma;
    double phi = std::exp(-0.5 * z * z) / std::sqrt(2 * M_PI);
    double Phi = 0.5 * std::erfc(-z / std::sqrt(2.0));
    double ei = (mu - fbest - 0.01) * Phi + sigma * phi;
    if (ei > best_ei) {
      best_ei = ei;
      best_x = x;
    }
  }
  return Denormalize(best_x);
}

}  // namespace hvd
