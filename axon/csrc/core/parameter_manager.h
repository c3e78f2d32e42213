// Online autotuner for the two engine knobs: fusion threshold (bytes) and
// cycle time (ms) — capability parity with the reference's ParameterManager
// + BayesianOptimization (SURVEY.md §2.1 "Parameter manager (autotuner)"),
// scored end-to-end by optimizer steps/sec (reported by
// DistributedOptimizer.step via autotune_step_mark; falls back to
// allreduced bytes/sec for op-only workloads). Three phases:
//   1. coarse 4x3 grid over (threshold, cycle);
//   2. AXON_AUTOTUNE_BO (default 6) Bayesian-optimization rounds — a GP
//      with an RBF kernel over (log2 threshold, log2 cycle) fits all
//      observations and the expected-improvement maximum over a
//      half-octave lattice is sampled next;
//   3. optional legacy 3x3 refinement (AXON_AUTOTUNE_REFINE) around the
//      incumbent.
//
// Enable with AXON_AUTOTUNE=1; AXON_AUTOTUNE_LOG=1 logs every sample.
// Runs on rank 0 only: the fusion threshold is a coordinator decision, and
// cycle-time updates ship to workers inside the ResponseList.
#pragma once

#include <chrono>
#include <cstdint>
#include <vector>

namespace axon {

class ParameterManager {
 public:
  ParameterManager(int64_t threshold, double cycle_ms);

  bool enabled() const { return enabled_; }
  // Account bytes moved by GPU allreduce responses this cycle.
  void record_bytes(int64_t bytes);
  // Account optimizer steps completed since the last cycle (end-to-end
  // signal: steps/sec outranks allreduced-bytes/sec when present).
  void record_steps(int64_t steps);
  // Called once per cycle on the coordinator. Returns true when the knobs
  // changed; outputs the new values (to apply locally + broadcast).
  bool maybe_update(int64_t* threshold, double* cycle_ms);

  struct Sample {
    int64_t threshold;
    double cycle_ms;
    double score = -1.0;
  };

 private:
  void advance();

  void refine();  // build a finer plan around the current best

  bool enabled_;
  bool log_;
  bool done_ = false;
  int bo_rounds_ = 6;
  int bo_done_ = 0;
  std::vector<Sample> history_;  // every scored sample (GP observations)
  int64_t bytes_ = 0;
  int64_t steps_ = 0;
  bool using_steps_ = false;
  int refine_rounds_ = 1;
  int round_ = 0;
  std::chrono::steady_clock::time_point window_start_;
  double warmup_until_ = 3.0;  // seconds before sampling (AXON_AUTOTUNE_WARMUP_SEC)
  double sample_sec_;
  std::chrono::steady_clock::time_point t0_;
  std::vector<Sample> plan_;
  size_t idx_ = 0;
  Sample best_;
  bool started_ = false;
};

// Bayesian-optimization proposal (exposed for deterministic unit tests):
// fit a fixed-hyperparameter GP (RBF over normalized log2 coordinates) to
// the scored samples and return the expected-improvement maximizer over a
// half-octave (threshold, cycle) lattice that is not already observed.
// Returns false when every lattice point has been observed or inputs are
// degenerate.
bool autotune_bo_propose(const std::vector<ParameterManager::Sample>& obs,
                         int64_t* threshold, double* cycle_ms);

}  // namespace axon
