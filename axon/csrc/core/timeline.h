// Chrome-trace (about:tracing) timeline writer — capability parity with the
// reference engine's Timeline: per-tensor lanes with
// NEGOTIATE / QUEUE / op-phase spans. Enabled by AXON_TIMELINE=/path.json.
#pragma once

#include <chrono>
#include <cstdio>
#include <mutex>
#include <string>
#include <unordered_map>

namespace axon {

class Timeline {
 public:
  void init(int rank);
  ~Timeline();
  bool enabled() const { return f_ != nullptr; }

  // Runtime control (hvd.start_timeline/stop_timeline parity): open a new
  // trace file (closing any active one) / close the active one. Safe to
  // call from the user thread while the background thread records spans.
  void start(const std::string& path);
  void stop();

  // Begin/end a named phase on the tensor's lane.
  void begin(const std::string& tensor, const char* phase);
  void end(const std::string& tensor);
  // Instant marker on the cycle lane (AXON_TIMELINE_MARK_CYCLES=1).
  void mark_cycle();

 private:
  int64_t now_us() const;
  int lane(const std::string& tensor);
  void emit(const char* ph, const char* name, int tid, int64_t ts);

  FILE* f_ = nullptr;
  int rank_ = 0;
  bool mark_cycles_ = false;
  std::mutex mu_;
  std::unordered_map<std::string, int> lanes_;
  std::unordered_map<std::string, std::string> open_phase_;
  int next_lane_ = 1;
  std::chrono::steady_clock::time_point t0_;
};

}  // namespace axon
