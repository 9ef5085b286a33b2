// Common device helpers for the maskrcnn_amd CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

// Ceil-div.
__host__ __device__ inline int ceil_div(int a, int b) {
  return (a + b - 1) / b;
}

// dtype load/store helpers: kernels compute in fp32, IO in T.
template <typename T> __device__ inline float ld(const T* p) {
  return static_cast<float>(*p);
}
template <> __device__ inline float ld<__hip_bfloat16>(const __hip_bfloat16* p) {
  return __bfloat162float(*p);
}
template <typename T> __device__ inline void st(T* p, float v) {
  *p = static_cast<T>(v);
}
template <> __device__ inline void st<__hip_bfloat16>(__hip_bfloat16* p, float v) {
  *p = __float2bfloat16(v);
}

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,    \
             __LINE__);                                                     \
    }                                                                       \
  } while (0)
