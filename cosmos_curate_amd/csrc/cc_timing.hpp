// Kernel-launch timing helpers (HIP side; see cc_common.hpp TimingState).
#pragma once

#include <hip/hip_runtime.h>

#include <vector>

#include "cc_common.hpp"

namespace cc {

// begin a timed launch: record ev0 on the stream.  Returns false when
// timing is disabled (no events created).
inline bool timed_begin(uint64_t stream, hipEvent_t* ev0, hipEvent_t* ev1) {
  auto& ts = timing();
  if (!ts.enabled) return false;
  if (hipEventCreate(ev0) != hipSuccess) return false;
  if (hipEventCreate(ev1) != hipSuccess) {
    hipEventDestroy(*ev0);
    return false;
  }
  hipEventRecord(*ev0, (hipStream_t)stream);
  return true;
}

// end a timed launch: record ev1 and queue the pair (drained at report).
inline void timed_end(const char* name, uint64_t stream, hipEvent_t ev0,
                      hipEvent_t ev1) {
  hipEventRecord(ev1, (hipStream_t)stream);
  auto& ts = timing();
  std::lock_guard<std::mutex> lk(ts.mu);
  ts.pending.push_back({name, (void*)ev0, (void*)ev1});
}

// drain pending pairs into the per-kernel totals (synchronizes on the
// recorded events; call from cc_timing_report / cc_timing_reset).
inline void timed_drain() {
  auto& ts = timing();
  std::vector<PendingPair> pend;
  {
    std::lock_guard<std::mutex> lk(ts.mu);
    pend.swap(ts.pending);
  }
  for (auto& p : pend) {
    hipEventSynchronize((hipEvent_t)p.ev1);
    float ms = 0;
    hipEventElapsedTime(&ms, (hipEvent_t)p.ev0, (hipEvent_t)p.ev1);
    {
      std::lock_guard<std::mutex> lk(ts.mu);
      auto& e = ts.entries[p.name];
      e.total_ms += ms;
      e.count += 1;
    }
    hipEventDestroy((hipEvent_t)p.ev0);
    hipEventDestroy((hipEvent_t)p.ev1);
  }
}

}  // namespace cc
