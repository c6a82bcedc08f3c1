// Common helpers for flowhip gfx950 kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define FLOWHIP_CHECK_HIP(expr)                                              \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,     \
             __LINE__);                                                      \
      abort();                                                               \
    }                                                                        \
  } while (0)

// ceil-div
__host__ __device__ static inline int fh_cdiv(int a, int b) { return (a + b - 1) / b; }

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;   // 4 VGPRs
using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;
