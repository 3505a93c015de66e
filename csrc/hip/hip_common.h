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

// ---- fp8 (OCP e4m3) KV-cache element support ------------------------------
// gfx950 hardware converts: v_cvt_f32_fp8 unpacks one byte of a dword
// (constant byte select), v_cvt_pk_fp8_f32 packs two floats into half a
// dword. Cache element type CT is short (bf16) or unsigned char (fp8).
typedef __attribute__((ext_vector_type(8))) unsigned char uchar8;

// load 8 cache elements -> f32[8]
__device__ __forceinline__ void load_kv8(const short* p, float* f) {
  short8 v = *(const short8*)p;
#pragma unroll
  for (int j = 0; j < 8; ++j) f[j] = bf16_to_f32(v[j]);
}

__device__ __forceinline__ void load_kv8(const unsigned char* p, float* f) {
  const uint32_t* w = (const uint32_t*)p;   // two dwords = 8 fp8
  uint32_t a = w[0], b = w[1];
  f[0] = __builtin_amdgcn_cvt_f32_fp8(a, 0);
  f[1] = __builtin_amdgcn_cvt_f32_fp8(a, 1);
  f[2] = __builtin_amdgcn_cvt_f32_fp8(a, 2);
  f[3] = __builtin_amdgcn_cvt_f32_fp8(a, 3);
  f[4] = __builtin_amdgcn_cvt_f32_fp8(b, 0);
  f[5] = __builtin_amdgcn_cvt_f32_fp8(b, 1);
  f[6] = __builtin_amdgcn_cvt_f32_fp8(b, 2);
  f[7] = __builtin_amdgcn_cvt_f32_fp8(b, 3);
}

// load 2 consecutive cache elements (phase-C dim pair per lane)
__device__ __forceinline__ void load_kv2(const short* p, float& a, float& b) {
  const int32_t pair = *(const int32_t*)p;
  a = bf16_to_f32((short)(pair & 0xFFFF));
  b = bf16_to_f32((short)((pair >> 16) & 0xFFFF));
}

__device__ __forceinline__ void load_kv2(const unsigned char* p, float& a,
                                         float& b) {
  const uint32_t pair = *(const uint16_t*)p;
  a = __builtin_amdgcn_cvt_f32_fp8(pair, 0);
  b = __builtin_amdgcn_cvt_f32_fp8(pair, 1);
}

// store 8 bf16 source elements as cache elements
__device__ __forceinline__ void store_kv8(short* p, short8 v) {
  *(short8*)p = v;
}

__device__ __forceinline__ void store_kv8(unsigned char* p, short8 v) {
  uint32_t lo = 0, hi = 0;
  lo = __builtin_amdgcn_cvt_pk_fp8_f32(bf16_to_f32(v[0]), bf16_to_f32(v[1]),
                                       lo, false);
  lo = __builtin_amdgcn_cvt_pk_fp8_f32(bf16_to_f32(v[2]), bf16_to_f32(v[3]),
                                       lo, true);
  hi = __builtin_amdgcn_cvt_pk_fp8_f32(bf16_to_f32(v[4]), bf16_to_f32(v[5]),
                                       hi, false);
  hi = __builtin_amdgcn_cvt_pk_fp8_f32(bf16_to_f32(v[6]), bf16_to_f32(v[7]),
                                       hi, true);
  uint32_t* w = (uint32_t*)p;
  w[0] = lo;
  w[1] = hi;
}

// load 8 cache elements as bf16 (LDS staging in the flash-prefill kernel)
__device__ __forceinline__ short8 load_kv8_bf16(const short* p) {
  return *(const short8*)p;
}

__device__ __forceinline__ short8 load_kv8_bf16(const unsigned char* p) {
  float f[8];
  load_kv8(p, f);
  short8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = f32_to_bf16(f[j]);
  return out;
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
