// Shared infrastructure for libcchot.so: error state + kernel timing.
#pragma once

#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <map>
#include <mutex>
#include <string>

#include "../../include/cc_hotpath.h"

namespace cc {

// thread-local last error message
inline thread_local std::string g_last_error;

inline int set_error(int code, const char* fmt, ...) {
  char buf[512];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  g_last_error = buf;
  return code;
}

// ---- kernel timing registry (hipEvent based, see cc_timing_* ABI) ----
// Launches are bracketed with event pairs on their stream but NOT
// synchronized at launch time (that would serialize the pipeline and
// time isolated launches); pending pairs drain at cc_timing_report.
struct TimingEntry {
  double total_ms = 0.0;
  int64_t count = 0;
};

struct PendingPair {
  std::string name;
  void* ev0;
  void* ev1;
};

struct TimingState {
  std::mutex mu;
  bool enabled = false;
  std::map<std::string, TimingEntry> entries;
  std::vector<PendingPair> pending;
};

inline TimingState& timing() {
  static TimingState s;
  return s;
}

}  // namespace cc
