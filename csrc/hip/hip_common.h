// hip_common.h — shared device helpers for the gfx950 (CDNA4) kernels.
// Wave width is 64 on CDNA4; every cross-lane idiom below assumes that.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define WAVE 64

// vectorized bf16 access: 8 bf16 = 16 B per lane (coalescing sweet spot, guide G13)
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(2))) float float2v;

__device__ __forceinline__ float bf16_to_f32(short u) {
  union { float f; uint32_t i; } c;
  c.i = ((uint32_t)(uint16_t)u) << 16;
  return c.f;
}

__device__ __forceinline__ short f32_to_bf16(float f) {
  union { float f; uint32_t i; } c;
  c.f = f;
  // round-to-nearest-even
  uint32_t lsb = (c.i >> 16) & 1;
  uint32_t rounded = c.i + 0x7FFF + lsb;
  return (short)(uint16_t)(rounded >> 16);
}

// full-wave f32 sum reduction (6 xor-shuffle steps over 64 lanes)
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// block-level f32 sum reduction across NW waves (NW <= 16), via LDS scratch.
// `scratch` must hold NW floats; returns the total to all threads.
template <int NW>
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  int wave = threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wave] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < NW; ++w) total += scratch[w];
  return total;
}

#define HIP_CHECK_LAST()                                              \
  do {                                                                 \
    hipError_t e_ = hipGetLastError();                                 \
    if (e_ != hipSuccess) {                                            \
      return e_;                                                       \
    }                                                                  \
  } while (0)
