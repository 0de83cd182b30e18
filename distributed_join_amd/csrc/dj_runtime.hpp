/* dj_runtime.hpp — internal: streams shared between the C ABI and the C++
 * orchestration layer. Implemented in dj_capi.hip. */
#pragma once
#include <hip/hip_runtime.h>
hipStream_t dj_rt_stream();       // compute stream
hipStream_t dj_rt_comm_stream();  // communication stream
