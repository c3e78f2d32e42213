// Core runtime: global state + the background operation loop (the analog of
// horovod/common/operations.cc BackgroundThreadLoop/RunLoopOnce +
// global_state.h). One background thread per process drains
// the tensor queue every cycle, coordinates via the Controller, and
// dispatches responses to the CPU or RCCL backend.
#pragma once

#include "common.h"

namespace axon {

// Start the core: spawns the background thread, connects the control plane.
// control_path: unix socket path for the single-node rendezvous.
// Blocks until the control plane is connected on all ranks (or throws).
void initialize(int rank, int size, int local_rank, const std::string& control_path);

// Graceful shutdown: rendezvous on an internal barrier, stop the loop.
// Blocks until the background thread exits.
void shutdown();

bool is_initialized();
int rank();
int size();
int local_rank();

// Enqueue a collective. entry.callback fires from a core thread when the op
// completes (never holds the GIL). Throws if not initialized or the loop
// died (fatal peer failure).
void enqueue(TensorEntry entry);

// Non-empty when the background loop died unexpectedly (peer failure).
std::string fatal_error();

// Process-set registry (SURVEY.md §2.1 "Process sets"). Id 0 (the global
// set) is created by initialize(). ps_register must be called collectively
// and in identical order on every rank (the Python API enforces this with
// an allgather of the member list); it returns the same id everywhere.
int ps_register(const std::vector<int>& ranks);
ProcessSetInfo ps_get(int id);  // copy; unknown id -> invalid info (valid=false, empty)
bool ps_known(int id);
void ps_remove(int id);  // marks invalid; ops on it then error
int ps_count();
int ps_version();  // bumped on register and remove; controller rescan key

// End-to-end autotune signal: called once per optimizer step (rank-local);
// rank 0's marks feed the ParameterManager's steps/sec score.
void pm_step_mark();

// Runtime timeline control (hvd.start_timeline/stop_timeline parity);
// rank-local, any time after initialize().
void timeline_start(const std::string& path);
void timeline_stop();

// GPU backend factory seam (implemented in rccl_backend.cpp).
class ControlPlane;
class Timeline;
class GpuBackend {
 public:
  virtual ~GpuBackend() = default;
  // Execute asynchronously on the comm stream; fire entry callbacks from
  // the completion-poller thread. Takes ownership of entries.
  virtual void execute(const struct Response& r, std::vector<TensorEntry> entries) = 0;
  // Drain outstanding work, stop the poller.
  virtual void finalize() = 0;
};
GpuBackend* create_rccl_backend(int rank, int size, ControlPlane* cp, Timeline* tl);

}  // namespace axon
