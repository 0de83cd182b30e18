/* dj_timing.hpp — hipEvent phase-timing registry shared by the C ABI and the
 * C++ orchestration layer (replaces the reference's report_timing wall-clock
 * prints, distributed_join.cpp:120-130). Implemented in dj_capi.hip. */
#pragma once

#include <hip/hip_runtime.h>

namespace dj_timing {

bool enabled();
void record_begin(int phase, hipStream_t s);
void record_end(hipStream_t s);

struct Scope {
  hipStream_t s;
  bool on;
  Scope(int phase, hipStream_t st) : s(st), on(enabled())
  {
    if (on) record_begin(phase, st);
  }
  ~Scope()
  {
    if (on) record_end(s);
  }
};

}  // namespace dj_timing
