/*
 * dj_error.hpp — error macros for the MI355X-native distributed join.
 * Mirrors the behavior of the reference's error.hpp:22-99 (print + exit(1))
 * with HIP/RCCL equivalents of CUDA_RT_CALL / NCCL_CALL.
 */
#pragma once

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

#define DJ_HIP_CALL(call)                                                              \
  do {                                                                                 \
    hipError_t _status = (call);                                                       \
    if (_status != hipSuccess) {                                                       \
      fprintf(stderr,                                                                  \
              "ERROR: HIP call \"%s\" in line %d of file %s failed with %s (%d).\n",   \
              #call, __LINE__, __FILE__, hipGetErrorString(_status), _status);         \
      exit(1);                                                                         \
    }                                                                                  \
  } while (0)

#define DJ_CHECK_ERROR(expr, msg)                                                      \
  do {                                                                                 \
    if (!(expr)) {                                                                     \
      fprintf(stderr, "ERROR: %s (line %d of %s)\n", msg, __LINE__, __FILE__);         \
      exit(1);                                                                         \
    }                                                                                  \
  } while (0)

#define DJ_RCCL_CALL(call)                                                             \
  do {                                                                                 \
    ncclResult_t _status = (call);                                                     \
    if (_status != ncclSuccess) {                                                      \
      fprintf(stderr,                                                                  \
              "ERROR: RCCL call \"%s\" in line %d of file %s failed with %s (%d).\n",  \
              #call, __LINE__, __FILE__, ncclGetErrorString(_status), _status);        \
      exit(1);                                                                         \
    }                                                                                  \
  } while (0)
