// Common helpers for ddlw CDNA4 (gfx950) kernels.
// Layout convention: activations are NHWC bf16 (torch channels_last);
// statistics/parameters are fp32. All reductions accumulate in fp32.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DDLW_EXPORT extern "C" __attribute__((visibility("default")))

typedef unsigned short bf16_t;
typedef unsigned int u32;
typedef unsigned long long u64;

// 8 bf16 = one 16-byte vector load (guideline: vectorize bf16 as short8)
union bf16x8 {
  uint4 v;
  bf16_t h[8];
};

__device__ __forceinline__ float b2f(bf16_t h) {
  union { float f; u32 u; } cvt;
  cvt.u = ((u32)h) << 16;
  return cvt.f;
}

__device__ __forceinline__ bf16_t f2b(float f) {
  // hardware RNE convert (v_cvt path; same rounding as the manual
  // integer-bias form but ~1 VALU instead of ~5)
  __bf16 h = (__bf16)f;
  union { __bf16 h; bf16_t u; } cvt;
  cvt.h = h;
  return cvt.u;
}

__device__ __forceinline__ float warp_reduce_sum(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

__device__ __forceinline__ float warp_reduce_max(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// error plumbing ------------------------------------------------------------
#define DDLW_CHECK_LAUNCH()                                                    \
  do {                                                                         \
    hipError_t err_ = hipGetLastError();                                       \
    if (err_ != hipSuccess) {                                                  \
      ddlw_set_error(hipGetErrorString(err_));                                 \
      return 1;                                                                \
    }                                                                          \
    return 0;                                                                  \
  } while (0)

void ddlw_set_error(const char* msg);
