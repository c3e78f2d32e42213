#include "timeline.h"

#include "common.h"

namespace axon {

void Timeline::init(int rank) {
  rank_ = rank;
  std::string path = env_str("AXON_TIMELINE", "");
  if (path.empty()) return;
  start(path);
}

void Timeline::start(const std::string& path) {
  std::lock_guard<std::mutex> lk(mu_);
  if (f_) {
    std::fclose(f_);
    f_ = nullptr;
    open_phase_.clear();
  }
  // One file per rank (single writer): path with rank suffix unless ws==1
  // callers pass rank-qualified path already via launcher env.
  std::string p = path;
  auto dot = p.rfind(".json");
  std::string suffix = "." + std::to_string(rank_);
  if (dot != std::string::npos)
    p = p.substr(0, dot) + suffix + ".json";
  else
    p += suffix;
  f_ = std::fopen(p.c_str(), "w");
  if (!f_) {
    AXON_LOG(LogLevel::WARN, "axon timeline: cannot open " + p);
    return;
  }
  mark_cycles_ = env_int("AXON_TIMELINE_MARK_CYCLES", 0) != 0;
  t0_ = std::chrono::steady_clock::now();
  std::fprintf(f_, "[\n");
  std::fprintf(f_,
               "{\"name\":\"process_name\",\"ph\":\"M\",\"pid\":%d,"
               "\"args\":{\"name\":\"axon rank %d\"}},\n",
               rank_, rank_);
}

void Timeline::stop() {
  std::lock_guard<std::mutex> lk(mu_);
  if (!f_) return;
  // Close any spans left open so the trace stays well-formed.
  for (auto& kv : open_phase_)
    if (!kv.second.empty()) emit("E", kv.second.c_str(), lanes_[kv.first], now_us());
  open_phase_.clear();
  std::fclose(f_);
  f_ = nullptr;
}

Timeline::~Timeline() {
  if (f_) {
    // Trailing comma is accepted by chrome://tracing and perfetto.
    std::fclose(f_);
  }
}

int64_t Timeline::now_us() const {
  return std::chrono::duration_cast<std::chrono::microseconds>(
             std::chrono::steady_clock::now() - t0_)
      .count();
}

int Timeline::lane(const std::string& tensor) {
  auto it = lanes_.find(tensor);
  if (it != lanes_.end()) return it->second;
  int id = next_lane_++;
  lanes_[tensor] = id;
  std::fprintf(f_,
               "{\"name\":\"thread_name\",\"ph\":\"M\",\"pid\":%d,\"tid\":%d,"
               "\"args\":{\"name\":\"%s\"}},\n",
               rank_, id, tensor.c_str());
  return id;
}

void Timeline::emit(const char* ph, const char* name, int tid, int64_t ts) {
  std::fprintf(f_, "{\"name\":\"%s\",\"ph\":\"%s\",\"pid\":%d,\"tid\":%d,\"ts\":%lld},\n",
               name, ph, rank_, tid, static_cast<long long>(ts));
}

void Timeline::begin(const std::string& tensor, const char* phase) {
  std::lock_guard<std::mutex> lk(mu_);
  if (!f_) return;
  int id = lane(tensor);
  auto& open = open_phase_[tensor];
  if (!open.empty()) emit("E", open.c_str(), id, now_us());
  open = phase;
  emit("B", phase, id, now_us());
}

void Timeline::end(const std::string& tensor) {
  std::lock_guard<std::mutex> lk(mu_);
  if (!f_) return;
  auto it = open_phase_.find(tensor);
  if (it == open_phase_.end() || it->second.empty()) return;
  emit("E", it->second.c_str(), lanes_[tensor], now_us());
  it->second.clear();
}

void Timeline::mark_cycle() {
  std::lock_guard<std::mutex> lk(mu_);
  if (!f_ || !mark_cycles_) return;
  std::fprintf(f_, "{\"name\":\"cycle\",\"ph\":\"i\",\"pid\":%d,\"tid\":0,\"ts\":%lld,\"s\":\"p\"},\n",
               rank_, static_cast<long long>(now_us()));
}

}  // namespace axon
