#include "parameter_manager.h"

#include <algorithm>
#include <cmath>

#include "common.h"

namespace axon {

namespace {

// ---- Bayesian-optimization machinery (fixed-hyperparameter GP + EI) ----
// Feature space: x1 = log2(threshold/MiB) in [3, 10] (8 MiB .. 1 GiB),
// x2 = log2(cycle_ms) in [-2, 2] (0.25 .. 4 ms), both normalized to [0,1].

constexpr double kL = 0.25;     // RBF length scale (normalized units)
constexpr double kSigF2 = 1.0;  // signal variance (y standardized)
constexpr double kSigN2 = 0.01; // noise variance
constexpr double kXi = 0.01;    // EI exploration margin

struct XY {
  double x1, x2;
};

double rbf(const XY& a, const XY& b) {
  double d1 = a.x1 - b.x1, d2 = a.x2 - b.x2;
  return kSigF2 * std::exp(-(d1 * d1 + d2 * d2) / (2.0 * kL * kL));
}

XY to_x(int64_t thr, double cyc) {
  const double MiB = 1 << 20;
  double l2t = std::log2(std::max<double>(1.0, thr / MiB));
  double l2c = std::log2(std::max(1e-3, cyc));
  return {(l2t - 3.0) / 7.0, (l2c + 2.0) / 4.0};
}

// In-place Cholesky of a small SPD matrix (row-major n x n); false if not
// positive definite.
bool cholesky(std::vector<double>& a, int n) {
  for (int i = 0; i < n; ++i) {
    for (int j = 0; j <= i; ++j) {
      double s = a[i * n + j];
      for (int k = 0; k < j; ++k) s -= a[i * n + k] * a[j * n + k];
      if (i == j) {
        if (s <= 0) return false;
        a[i * n + i] = std::sqrt(s);
      } else {
        a[i * n + j] = s / a[j * n + j];
      }
    }
  }
  return true;
}

// Solve L L^T x = b given the Cholesky factor in `a`'s lower triangle.
void chol_solve(const std::vector<double>& a, int n, std::vector<double>& b) {
  for (int i = 0; i < n; ++i) {  // forward: L y = b
    double s = b[i];
    for (int k = 0; k < i; ++k) s -= a[i * n + k] * b[k];
    b[i] = s / a[i * n + i];
  }
  for (int i = n - 1; i >= 0; --i) {  // backward: L^T x = y
    double s = b[i];
    for (int k = i + 1; k < n; ++k) s -= a[k * n + i] * b[k];
    b[i] = s / a[i * n + i];
  }
}

double norm_cdf(double z) { return 0.5 * std::erfc(-z / std::sqrt(2.0)); }
double norm_pdf(double z) {
  return std::exp(-0.5 * z * z) / std::sqrt(2.0 * M_PI);
}

}  // namespace

bool autotune_bo_propose(const std::vector<ParameterManager::Sample>& obs,
                         int64_t* threshold, double* cycle_ms) {
  const int n = static_cast<int>(obs.size());
  if (n < 2) return false;
  std::vector<XY> xs(n);
  std::vector<double> y(n);
  double mean = 0, var = 0;
  for (int i = 0; i < n; ++i) {
    xs[i] = to_x(obs[i].threshold, obs[i].cycle_ms);
    y[i] = obs[i].score;
    mean += y[i];
  }
  mean /= n;
  for (int i = 0; i < n; ++i) var += (y[i] - mean) * (y[i] - mean);
  double sd = std::sqrt(var / std::max(1, n - 1));
  if (!(sd > 0)) return false;  // flat scores: nothing to model
  for (int i = 0; i < n; ++i) y[i] = (y[i] - mean) / sd;
  double ybest = *std::max_element(y.begin(), y.end());

  std::vector<double> K(static_cast<size_t>(n) * n);
  for (int i = 0; i < n; ++i)
    for (int j = 0; j < n; ++j)
      K[i * n + j] = rbf(xs[i], xs[j]) + (i == j ? kSigN2 : 0.0);
  if (!cholesky(K, n)) return false;
  std::vector<double> alpha = y;
  chol_solve(K, n, alpha);

  // Half-octave candidate lattice.
  const int64_t MiB = 1 << 20;
  double best_ei = 0.0;
  bool found = false;
  for (double l2t = 3.0; l2t <= 10.0 + 1e-9; l2t += 0.5) {
    for (double l2c = -2.0; l2c <= 2.0 + 1e-9; l2c += 0.5) {
      int64_t thr = static_cast<int64_t>(std::llround(std::pow(2.0, l2t) * MiB));
      double cyc = std::pow(2.0, l2c);
      XY c = to_x(thr, cyc);
      bool seen = false;
      for (auto& x : xs) {
        double d1 = x.x1 - c.x1, d2 = x.x2 - c.x2;
        seen |= (d1 * d1 + d2 * d2) < 1e-6;
      }
      if (seen) continue;
      std::vector<double> ks(n);
      for (int i = 0; i < n; ++i) ks[i] = rbf(xs[i], c);
      double mu = 0;
      for (int i = 0; i < n; ++i) mu += ks[i] * alpha[i];
      std::vector<double> v = ks;
      chol_solve(K, n, v);
      double s2 = kSigF2;
      for (int i = 0; i < n; ++i) s2 -= ks[i] * v[i];
      double sig = std::sqrt(std::max(1e-12, s2));
      double z = (mu - ybest - kXi) / sig;
      double ei = (mu - ybest - kXi) * norm_cdf(z) + sig * norm_pdf(z);
      if (ei > best_ei) {
        best_ei = ei;
        *threshold = thr;
        *cycle_ms = cyc;
        found = true;
      }
    }
  }
  return found;
}

ParameterManager::ParameterManager(int64_t threshold, double cycle_ms) {
  enabled_ = env_int("AXON_AUTOTUNE", 0) != 0;
  log_ = env_int("AXON_AUTOTUNE_LOG", 0) != 0;
  sample_sec_ = env_double("AXON_AUTOTUNE_SAMPLE_SEC", 1.0);
  warmup_until_ = env_double("AXON_AUTOTUNE_WARMUP_SEC", 3.0);
  refine_rounds_ = static_cast<int>(env_int("AXON_AUTOTUNE_REFINE", 1));
  bo_rounds_ = static_cast<int>(env_int("AXON_AUTOTUNE_BO", 6));
  best_ = Sample{threshold, cycle_ms, -1.0};
  if (!enabled_) return;
  // Coarse 2-D neighborhood around the xGMI-tuned defaults. Thresholds in
  // MiB x cycle times in ms; 12 samples ~= 12 s of tuning.
  const int64_t MiB = 1 << 20;
  for (int64_t thr : {32 * MiB, 64 * MiB, 128 * MiB, 256 * MiB}) {
    for (double cyc : {0.5, 1.0, 2.0}) {
      plan_.push_back(Sample{thr, cyc, -1.0});
    }
  }
  t0_ = std::chrono::steady_clock::now();
}

void ParameterManager::record_bytes(int64_t bytes) {
  if (enabled_ && !done_) bytes_ += bytes;
}

void ParameterManager::record_steps(int64_t steps) {
  if (enabled_ && !done_) steps_ += steps;
}

void ParameterManager::refine() {
  // One finer 3x3 neighborhood around the best sample so far (threshold
  // x0.5/x1/x2 clamped to [8 MiB, 1 GiB]; cycle x0.7/x1/x1.4), skipping
  // points already scored.
  const int64_t MiB = 1 << 20;
  std::vector<Sample> scored = history_;
  plan_.clear();
  idx_ = 0;
  for (double tf : {0.5, 1.0, 2.0}) {
    int64_t thr = static_cast<int64_t>(best_.threshold * tf);
    thr = std::max<int64_t>(8 * MiB, std::min<int64_t>(thr, 1024 * MiB));
    for (double cf : {0.7, 1.0, 1.4}) {
      double cyc = best_.cycle_ms * cf;
      bool seen = false;
      for (auto& s : scored)
        seen |= (s.threshold == thr && std::abs(s.cycle_ms - cyc) < 1e-9);
      for (auto& s : plan_)
        seen |= (s.threshold == thr && std::abs(s.cycle_ms - cyc) < 1e-9);
      if (!seen) plan_.push_back(Sample{thr, cyc, -1.0});
    }
  }
  // Re-seed with the incumbent so the final pick can keep it.
  plan_.push_back(best_);
}

bool ParameterManager::maybe_update(int64_t* threshold, double* cycle_ms) {
  if (!enabled_ || done_) return false;
  auto now = std::chrono::steady_clock::now();
  double since_start = std::chrono::duration<double>(now - t0_).count();
  if (since_start < warmup_until_) return false;

  if (!started_) {
    started_ = true;
    window_start_ = now;
    bytes_ = 0;
    *threshold = plan_[0].threshold;
    *cycle_ms = plan_[0].cycle_ms;
    return true;
  }
  double w = std::chrono::duration<double>(now - window_start_).count();
  if (w < sample_sec_) return false;
  if (steps_ > 0) using_steps_ = true;
  const int64_t signal = using_steps_ ? steps_ : bytes_;
  if (signal == 0) {  // idle window (no training yet) — restart the window
    window_start_ = now;
    bytes_ = 0;
    steps_ = 0;
    return false;
  }
  double score = static_cast<double>(signal) / w;
  plan_[idx_].score = score;
  history_.push_back(plan_[idx_]);
  if (log_)
    AXON_LOG(LogLevel::INFO,
             "autotune sample: threshold=" +
                 std::to_string(plan_[idx_].threshold >> 20) + "MiB cycle=" +
                 std::to_string(plan_[idx_].cycle_ms) + "ms -> " +
                 (using_steps_ ? std::to_string(score) + " steps/s"
                               : std::to_string(score / 1e9) + " GB/s"));
  bytes_ = 0;
  steps_ = 0;
  window_start_ = now;
  ++idx_;
  if (idx_ >= plan_.size()) {
    auto it = std::max_element(history_.begin(), history_.end(),
                               [](const Sample& a, const Sample& b) {
                                 return a.score < b.score;
                               });
    best_ = *it;
    // Phase 2: Bayesian-optimization rounds — one EI-maximizing proposal
    // at a time, fit on every observation so far.
    if (bo_done_ < bo_rounds_) {
      int64_t thr;
      double cyc;
      if (autotune_bo_propose(history_, &thr, &cyc)) {
        ++bo_done_;
        plan_ = {Sample{thr, cyc, -1.0}};
        idx_ = 0;
        if (log_)
          AXON_LOG(LogLevel::INFO,
                   "autotune BO proposal " + std::to_string(bo_done_) + "/" +
                       std::to_string(bo_rounds_) + ": threshold=" +
                       std::to_string(thr >> 20) + "MiB cycle=" +
                       std::to_string(cyc) + "ms");
        *threshold = thr;
        *cycle_ms = cyc;
        return true;
      }
      bo_done_ = bo_rounds_;  // degenerate scores / lattice exhausted
    }
    if (round_ < refine_rounds_) {
      ++round_;
      refine();
      if (!plan_.empty()) {
        *threshold = plan_[0].threshold;
        *cycle_ms = plan_[0].cycle_ms;
        return true;
      }
    }
    done_ = true;
    AXON_LOG(LogLevel::INFO,
             "autotune done: threshold=" + std::to_string(best_.threshold >> 20) +
                 "MiB cycle=" + std::to_string(best_.cycle_ms) + "ms");
    *threshold = best_.threshold;
    *cycle_ms = best_.cycle_ms;
    return true;
  }
  *threshold = plan_[idx_].threshold;
  *cycle_ms = plan_[idx_].cycle_ms;
  return true;
}

}  // namespace axon
