// Shared device helpers for petals_amd CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64

using short8 = __attribute__((ext_vector_type(8))) short;
using float4v = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
  union { float f; uint32_t i; } v;
  v.i = uint32_t(u) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
  union { float f; uint32_t i; } v;
  v.f = f;
  uint32_t lsb = (v.i >> 16) & 1;            // round-to-nearest-even
  uint32_t r = (v.i + 0x7FFFu + lsb) >> 16;
  return (unsigned short)r;
}

// full-wave fp32 sum (64 lanes)
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// sum across a 16-lane subgroup (lanes with equal lane/16)
__device__ __forceinline__ float group16_reduce_sum(float x) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e = hipGetLastError();                                        \
    if (e != hipSuccess) {                                                   \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e)); \
    }                                                                        \
  } while (0)

// finite stand-ins for -inf: the extension compiles with -ffast-math, under
// which IEEE inf comparisons/arithmetic are not reliable
#define NEG_SENTINEL  (-1e30f)
#define NEG_THRESHOLD (-1e29f)
