/* dj_runtime.hpp — internal: streams shared between the C ABI and the C++
 * orchestration layer. Implemented in dj_capi.hip. */
#pragma once
#include <hip/hip_runtime.h>
hipStream_t dj_rt_stream();       // compute stream
hipStream_t dj_rt_comm_stream();  // communication stream

/* enqueue-only bucketed join core (implemented in dj_capi.hip inside the
 * extern "C" block; used by the pipelined C++ orchestration) */
#include <cstdint>
extern "C" void dj_bucket_local_join_enqueue(const int64_t* d_lk, const int64_t* d_lp, int64_t ln,
                                  const int64_t* d_rk, const int64_t* d_rp, int64_t rn,
                                  int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                                  int64_t* d_out3, int64_t cap, int64_t* d_counter,
                                  int* d_error, int* d_any_overflow, void* d_scratch);
