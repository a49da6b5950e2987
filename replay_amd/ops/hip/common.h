// Common helpers for replay_amd gfx950 (CDNA4) kernels.
// Wave width is 64 on CDNA4 (MI355X): every warp idiom below is 64-wide.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

// Full-wave butterfly reduction (64 lanes).
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    v += __shfl_xor(v, off, WAVE);
  }
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    v = fmaxf(v, __shfl_xor(v, off, WAVE));
  }
  return v;
}

// dtype conversion helpers -------------------------------------------------
template <typename T>
__device__ __forceinline__ float to_f32(T x);
template <>
__device__ __forceinline__ float to_f32<float>(float x) { return x; }
template <>
__device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 x) {
  return __bfloat162float(x);
}
template <>
__device__ __forceinline__ float to_f32<__half>(__half x) { return __half2float(x); }

template <typename T>
__device__ __forceinline__ T from_f32(float x);
template <>
__device__ __forceinline__ float from_f32<float>(float x) { return x; }
template <>
__device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}
template <>
__device__ __forceinline__ __half from_f32<__half>(float x) { return __float2half(x); }

#define HIP_CHECK(expr)                                                          \
  do {                                                                           \
    hipError_t _e = (expr);                                                      \
    if (_e != hipSuccess) {                                                      \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));                  \
    }                                                                            \
  } while (0)
