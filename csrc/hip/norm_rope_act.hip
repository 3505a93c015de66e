// norm_rope_act.hip — fused memory-bound decoder-layer kernels for gfx950.
// All are HBM-bound: bf16 traffic is vectorized as short8 (16 B/lane,
// guide G13 — scalar bf16 is ~2x slower), fp32 math in registers.
//
// These replace what the reference delegates to vLLM inside its model-server
// pods (the router itself has no kernels); in the MI355X-native engine the
// per-GPU worker runs them directly.
#include "hip_common.h"

namespace {

// ---------------------------------------------------------------------------
// rmsnorm (+ optional fused residual add):
//   if residual: r = r + x  (written back);  src = r
//   else:        src = x
//   y = src * rsqrt(mean(src^2) + eps) * w
// One block (256 threads) per token row; row length H % 8 == 0.
// Values are cached in registers between the sum-of-squares pass and the
// normalize pass (up to CACHE_CHUNKS*8 elems/thread), avoiding a second HBM
// read for H <= 256*CACHE_CHUNKS*8 (= 16384 at CACHE_CHUNKS=8).
// ---------------------------------------------------------------------------
template <bool RESIDUAL>
__global__ void rmsnorm_kernel(const short* __restrict__ x,
                               short* __restrict__ residual,
                               const short* __restrict__ w,
                               short* __restrict__ y, int H, float eps) {
  constexpr int CACHE_CHUNKS = 8;
  __shared__ float red[16];
  const int64_t row = blockIdx.x;
  const int tid = threadIdx.x;
  const int nthreads = blockDim.x;
  const short8* x8 = (const short8*)(x + row * H);
  short8* r8 = RESIDUAL ? (short8*)(residual + row * H) : nullptr;
  const short8* w8 = (const short8*)w;
  short8* y8 = (short8*)(y + row * H);
  const int nchunks = H / 8;

  float vals[CACHE_CHUNKS][8];
  float ss = 0.f;
  int ci = 0;
  for (int c = tid; c < nchunks; c += nthreads, ++ci) {
    short8 v = x8[c];
    float f[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) f[j] = bf16_to_f32(v[j]);
    if constexpr (RESIDUAL) {
      short8 r = r8[c];
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] += bf16_to_f32(r[j]);
      short8 ro;
#pragma unroll
      for (int j = 0; j < 8; ++j) ro[j] = f32_to_bf16(f[j]);
      r8[c] = ro;  // residual stream updated in place
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += f[j] * f[j];
    if (ci < CACHE_CHUNKS) {
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[ci][j] = f[j];
    }
  }
  float total = block_reduce_sum<4>(ss, red);
  const float inv = rsqrtf(total / (float)H + eps);

  ci = 0;
  for (int c = tid; c < nchunks; c += nthreads, ++ci) {
    float f[8];
    if (ci < CACHE_CHUNKS) {
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] = vals[ci][j];
    } else {
      // overflow path for very large H: re-read (residual already folded in)
      short8 v = RESIDUAL ? r8[c] : x8[c];
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] = bf16_to_f32(v[j]);
    }
    short8 wv = w8[c];
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f32_to_bf16(f[j] * inv * bf16_to_f32(wv[j]));
    y8[c] = o;
  }
}

// ---------------------------------------------------------------------------
// RoPE (Llama/NeoX rotate-half), in place on q and k.
//   q: [T, n_q_heads, D], k: [T, n_kv_heads, D], D = 2*half
//   cos_sin: [max_pos, half, 2] f32 precomputed on host (guide App. B: no
//   on-device trig), positions: [T] int32
// grid = (T, n_q_heads + n_kv_heads); one wave per (token, head);
// lane i < half handles the (i, i+half) pair.
// ---------------------------------------------------------------------------
__global__ void rope_kernel(short* __restrict__ q, short* __restrict__ k,
                            const float* __restrict__ cos_sin,
                            const int32_t* __restrict__ positions, int n_tokens,
                            int n_q_heads, int n_kv_heads, int D) {
  const int t = blockIdx.x;
  const int head = blockIdx.y;
  const int half = D / 2;
  const int lane = threadIdx.x;
  if (t >= n_tokens || lane >= half) return;
  short* base = (head < n_q_heads)
                    ? q + ((int64_t)t * n_q_heads + head) * D
                    : k + ((int64_t)t * n_kv_heads + (head - n_q_heads)) * D;
  const float* cs = cos_sin + ((int64_t)positions[t] * half + lane) * 2;
  const float c = cs[0], s = cs[1];
  float x1 = bf16_to_f32(base[lane]);
  float x2 = bf16_to_f32(base[lane + half]);
  base[lane] = f32_to_bf16(x1 * c - x2 * s);
  base[lane + half] = f32_to_bf16(x2 * c + x1 * s);
}

// ---------------------------------------------------------------------------
// silu_mul: y = silu(gate) * up. gate/up are the two halves of one fused
// [T, 2*I] gate_up projection output (so the producing GEMM stays a single
// hipBLASLt call). Vectorized grid-stride over T rows.
// ---------------------------------------------------------------------------
__global__ void silu_mul_kernel(const short* __restrict__ gate_up,
                                short* __restrict__ y, int64_t T, int I) {
  const int nchunks = I / 8;
  const int64_t total = T * nchunks;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = i / nchunks;
    int c = (int)(i % nchunks);
    const short8* g8 = (const short8*)(gate_up + row * 2 * I);
    const short8* u8 = (const short8*)(gate_up + row * 2 * I + I);
    short8 g = g8[c], u = u8[c];
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32(g[j]);
      float uf = bf16_to_f32(u[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      o[j] = f32_to_bf16(gf * sig * uf);
    }
    ((short8*)(y + row * I))[c] = o;
  }
}

}  // namespace

extern "C" {

hipError_t lds_rmsnorm(const void* x, void* residual, const void* w, void* y,
                       int64_t n_rows, int H, float eps, hipStream_t stream) {
  if (n_rows == 0) return hipSuccess;
  dim3 grid((uint32_t)n_rows), block(256);
  if (residual) {
    hipLaunchKernelGGL(rmsnorm_kernel<true>, grid, block, 0, stream,
                       (const short*)x, (short*)residual, (const short*)w,
                       (short*)y, H, eps);
  } else {
    hipLaunchKernelGGL(rmsnorm_kernel<false>, grid, block, 0, stream,
                       (const short*)x, nullptr, (const short*)w, (short*)y, H,
                       eps);
  }
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lds_rope(void* q, void* k, const float* cos_sin,
                    const int32_t* positions, int n_tokens, int n_q_heads,
                    int n_kv_heads, int head_dim, hipStream_t stream) {
  if (n_tokens == 0) return hipSuccess;
  dim3 grid((uint32_t)n_tokens, (uint32_t)(n_q_heads + n_kv_heads));
  dim3 block(head_dim / 2 > 64 ? head_dim / 2 : 64);
  hipLaunchKernelGGL(rope_kernel, grid, block, 0, stream, (short*)q, (short*)k,
                     cos_sin, positions, n_tokens, n_q_heads, n_kv_heads,
                     head_dim);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lds_silu_mul(const void* gate_up, void* y, int64_t T, int I,
                        hipStream_t stream) {
  if (T == 0) return hipSuccess;
  int64_t total = T * (I / 8);
  int threads = 256;
  int blocks = (int)((total + threads - 1) / threads);
  if (blocks > 2048) blocks = 2048;  // grid-stride (guide G11)
  hipLaunchKernelGGL(silu_mul_kernel, dim3(blocks), dim3(threads), 0, stream,
                     (const short*)gate_up, (short*)y, T, I);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
