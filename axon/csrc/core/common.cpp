#include "common.h"

#include <cstdio>
#include <cstdlib>
#include <mutex>
#include <unistd.h>

namespace axon {

int64_t env_int(const char* name, int64_t dflt) {
  const char* v = std::getenv(name);
  if (!v || !*v) return dflt;
  return std::strtoll(v, nullptr, 10);
}

double env_double(const char* name, double dflt) {
  const char* v = std::getenv(name);
  if (!v || !*v) return dflt;
  return std::strtod(v, nullptr);
}

std::string env_str(const char* name, const std::string& dflt) {
  const char* v = std::getenv(name);
  if (!v || !*v) return dflt;
  return std::string(v);
}

LogLevel log_level() {
  static LogLevel lvl = [] {
    std::string s = env_str("AXON_LOG_LEVEL", "warn");
    if (s == "trace") return LogLevel::TRACE;
    if (s == "debug") return LogLevel::DEBUG;
    if (s == "info") return LogLevel::INFO;
    if (s == "warn" || s == "warning") return LogLevel::WARN;
    if (s == "error") return LogLevel::ERROR;
    if (s == "none" || s == "off") return LogLevel::NONE;
    return LogLevel::WARN;
  }();
  return lvl;
}

void log_msg(LogLevel lvl, const std::string& msg) {
  static std::mutex mu;
  static const char* names[] = {"TRACE", "DEBUG", "INFO", "WARN", "ERROR"};
  static int rank = static_cast<int>(env_int("AXON_RANK", env_int("RANK", -1)));
  std::lock_guard<std::mutex> lk(mu);
  std::fprintf(stderr, "[axon %s rank %d] %s\n",
               names[static_cast<int>(lvl)], rank, msg.c_str());
  std::fflush(stderr);
}

}  // namespace axon
