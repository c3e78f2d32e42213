// Coordination protocol messages: Request (worker -> rank 0) and Response
// (rank 0 -> all). Capability parity: horovod/common/message.h Request/
// Response wire protocol (SURVEY.md §2.1 "Controller / coordination"),
// re-designed as a compact single-node binary format.
#pragma once

#include "common.h"
#include "wire.h"

namespace axon {

struct Request {
  int32_t rank = 0;
  OpType op = OpType::ALLREDUCE;
  ReduceOp red = ReduceOp::AVERAGE;
  std::string name;
  uint8_t dtype = 0;
  uint8_t wire_dtype = 0;
  int32_t device = -1;  // -1 = CPU
  int32_t root_rank = 0;
  int32_t process_set = 0;  // process-set id (0 = global)
  std::vector<int64_t> shape;
  std::vector<int64_t> splits;  // alltoall: rows sent to each set member

  void serialize(Writer& w) const;
  static Request deserialize(Reader& r);
};

struct Response {
  OpType op = OpType::ALLREDUCE;
  ReduceOp red = ReduceOp::AVERAGE;
  std::vector<std::string> names;
  std::string error;  // non-empty => deliver error to callbacks
  uint8_t dtype = 0;
  uint8_t wire_dtype = 0;
  int32_t device = -1;
  int32_t root_rank = 0;
  int32_t process_set = 0;  // process-set id (0 = global)
  // Per-tensor total element count (fusion sizing; zero-fill for joined ranks).
  std::vector<int64_t> numels;
  // ALLGATHER: first_dims[t][r] = dim-0 extent contributed by rank r for
  // tensor t (variable-first-dim allgather).
  std::vector<std::vector<int64_t>> first_dims;
  // ALLTOALL: row-major [src_rank][dst_rank] row counts.
  std::vector<int64_t> splits_matrix;
  int32_t last_joined = -1;  // JOIN result
  // Response-cache ids assigned by the coordinator, parallel to `names`
  // (-1 = uncached). Workers re-submit steady-state allreduces as 4-byte
  // ids instead of full Requests (SURVEY.md §2.1 "Response cache").
  std::vector<int32_t> cache_ids;

  void serialize(Writer& w) const;
  static Response deserialize(Reader& r);
};

struct ResponseList {
  std::vector<Response> responses;
  bool shutdown = false;
  // Autotuner cycle-time update (ms); 0 = unchanged (parameter_manager.h).
  double new_cycle_ms = 0.0;

  std::vector<uint8_t> serialize() const;
  static ResponseList deserialize(const std::vector<uint8_t>& buf);
};

}  // namespace axon
