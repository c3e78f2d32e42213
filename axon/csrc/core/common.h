// axon core common definitions.
//
// Capability parity target: horovod/common/common.h (enums DataType,
// StatusType) per SURVEY.md §2.1 — re-designed for a torch-only,
// single-node, RCCL-only engine (no multi-backend dispatch).
#pragma once

#include <cstdint>
#include <functional>
#include <memory>
#include <string>
#include <vector>

#include <ATen/ATen.h>

namespace axon {

enum class OpType : uint8_t {
  ALLREDUCE = 0,
  ALLGATHER = 1,
  BROADCAST = 2,
  ALLTOALL = 3,
  REDUCESCATTER = 4,
  JOIN = 5,
  BARRIER = 6,
};

enum class ReduceOp : uint8_t {
  AVERAGE = 0,
  SUM = 1,
  ADASUM = 2,
  MIN = 3,
  MAX = 4,
  PRODUCT = 5,
};

enum class StatusType : uint8_t {
  OK = 0,
  ERROR = 1,
  ABORTED = 2,
  IN_PROGRESS = 3,
};

struct Status {
  StatusType type = StatusType::OK;
  std::string reason;
  static Status OK() { return Status{}; }
  static Status Error(std::string r) {
    return Status{StatusType::ERROR, std::move(r)};
  }
  static Status Aborted(std::string r) {
    return Status{StatusType::ABORTED, std::move(r)};
  }
  bool ok() const { return type == StatusType::OK; }
};

using StatusCallback = std::function<void(const Status&)>;

// Process set: a named subgroup of ranks with its own collectives and (on
// GPU) its own RCCL communicator (SURVEY.md §2.1 "Process sets"). Id 0 is
// the global set. Registration must happen collectively and in identical
// order on every rank (enforced by the Python API via allgather).
struct ProcessSetInfo {
  int id = 0;
  std::vector<int> ranks;  // sorted global ranks
  uint32_t mask = 0;       // bit r set iff rank r is a member
  int my_index = -1;       // this rank's position in `ranks`; -1 = non-member
  bool valid = true;       // false after remove_process_set
  int size() const { return static_cast<int>(ranks.size()); }
  bool member(int r) const { return (mask >> r) & 1u; }
  int index_of(int r) const {
    for (size_t i = 0; i < ranks.size(); ++i)
      if (ranks[i] == r) return static_cast<int>(i);
    return -1;
  }
};

// One pending collective operation on one rank.
struct TensorEntry {
  std::string name;
  OpType op = OpType::ALLREDUCE;
  ReduceOp red = ReduceOp::AVERAGE;
  at::Tensor in;    // input tensor (may be undefined for JOIN/BARRIER)
  at::Tensor out;   // output tensor (in-place ops: same as in)
  int root_rank = 0;          // broadcast root
  double prescale = 1.0;      // multiply before reduction
  double postscale = 1.0;     // multiply after reduction
  at::ScalarType wire_dtype = at::ScalarType::Undefined;  // fp16 compression
  std::vector<int64_t> splits;          // alltoall send splits (rows per rank)
  at::Tensor received_splits;           // alltoall: out tensor for recv splits
  int32_t device = -1;        // hip device index, -1 = CPU
  int32_t process_set = 0;    // process-set id (0 = global)
  uint64_t ready_event = 0;   // hipEvent_t recorded on producer stream (opaque)
  uint64_t handle = 0;        // binding-side handle id
  StatusCallback callback;
};

// Convert between at::ScalarType and a stable wire id.
inline uint8_t dtype_id(at::ScalarType t) { return static_cast<uint8_t>(t); }
inline at::ScalarType id_dtype(uint8_t v) { return static_cast<at::ScalarType>(v); }

inline size_t dtype_size(at::ScalarType t) { return at::elementSize(t); }

// Env helpers ---------------------------------------------------------------
int64_t env_int(const char* name, int64_t dflt);
double env_double(const char* name, double dflt);
std::string env_str(const char* name, const std::string& dflt);

// Logging -------------------------------------------------------------------
enum class LogLevel : int { TRACE = 0, DEBUG = 1, INFO = 2, WARN = 3, ERROR = 4, NONE = 5 };
LogLevel log_level();
void log_msg(LogLevel lvl, const std::string& msg);

#define AXON_LOG(lvl, msg)                                         \
  do {                                                             \
    if (static_cast<int>(lvl) >= static_cast<int>(::axon::log_level())) \
      ::axon::log_msg(lvl, msg);                                   \
  } while (0)

}  // namespace axon
