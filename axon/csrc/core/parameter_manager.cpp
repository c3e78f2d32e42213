#include "parameter_manager.h"

#include <algorithm>

#include "common.h"

namespace axon {

ParameterManager::ParameterManager(int64_t threshold, double cycle_ms) {
  enabled_ = env_int("AXON_AUTOTUNE", 0) != 0;
  log_ = env_int("AXON_AUTOTUNE_LOG", 0) != 0;
  sample_sec_ = env_double("AXON_AUTOTUNE_SAMPLE_SEC", 1.0);
  warmup_until_ = env_double("AXON_AUTOTUNE_WARMUP_SEC", 3.0);
  refine_rounds_ = static_cast<int>(env_int("AXON_AUTOTUNE_REFINE", 1));
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
  std::vector<Sample> scored = plan_;
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
    auto it = std::max_element(plan_.begin(), plan_.end(),
                               [](const Sample& a, const Sample& b) {
                                 return a.score < b.score;
                               });
    best_ = *it;
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
