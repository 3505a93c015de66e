// flash_prefill_glds.hip — the bf16-KV variant of the fused MFMA prefill
// attention with a glds staging pipeline (guide §5 "glds with >1 tile in
// flight"): K/V tiles are DMA'd straight to LDS (`global_load_lds`) into a
// 3-deep ring, one tile ahead, with COUNTED `s_waitcnt vmcnt(N)` and raw
// `s_barrier` (a `__syncthreads()` would drain the in-flight DMA; hipcc
// also drains glds at any ordinary in-loop global load, so the block table
// is staged into LDS up front and the hot loop touches global memory ONLY
// through glds). The LDS image stays lane-linear (a glds constraint); the
// XOR bank swizzle is applied by permuting the per-lane SOURCE addresses.
//
// Ring safety: iteration order is [counted vmcnt][raw barrier]
// [compute tile n][issue tile n+2] — barrier C_n separates every wave's
// last read of buffer (n-1)%3 (its compute at iteration n-1, before C_n)
// from the buffer's rewrite for tile n+2 (after C_n in every wave).
//
// fp8 KV cannot ride glds (raw byte copy, no conversion point), so the
// fp8 path keeps the convert-on-stage kernel in flash_prefill.hip; the
// launcher in ops.hip picks per cache dtype. Attention math, fragment
// layouts, masking and the swapped-QK^T softmax are identical to
// flash_prefill.hip (see its header comments).
#include "hip_common.h"

namespace {

constexpr int D = 128;
constexpr int KVBLK = 32;    // tokens per KV tile (ring granularity)
constexpr int NBUF = 3;
constexpr int NW = 4;
constexpr int QROWS = 32;
constexpr int BT_CAP = 1024;  // staged block-table entries (16K-token ctx)
constexpr float NEG = -1e30f;

constexpr int TILE_SHORTS = KVBLK * D;           // 4096 shorts = 8 KB
constexpr int GLDS_PER_WAVE = KVBLK * 16 / 64 / NW;  // K instrs/wave (=2)

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;
typedef __attribute__((ext_vector_type(4))) short short4v;

// Bank swizzle S(row): swap the two 2-bit halves of row&15. Chosen from
// the CDNA4 bank model (MI355X_MICROARCH.md "LDS"): the K ds_read_b128
// only needs S bijective on row&15 (its 16-lane groups cover all 16
// values of qcol&15, so any bijection spreads the 16 slots); the V
// ds_read_b64_tr_b16 (32-lane groups) additionally needs kv0&3 steered
// into address bits [7:6] so the 8-byte granule index
// (dt | g | r) ^ (S << 1) is injective over the group's (g, kv0, r) —
// the previous S(row)=row&15 left granule bits [4:3] untouched and put
// FOUR lanes on every granule (the measured 47%-of-LDS-active conflict
// cycles, profiles/r01_pmc_final.md).
__device__ __forceinline__ int swz4(int row) {
  return ((row & 3) << 2) | ((row >> 2) & 3);
}

__device__ __forceinline__ int g_swz(int row, int byte_off) {
  return row * 256 + (byte_off ^ (swz4(row) << 4));
}

__device__ __forceinline__ unsigned int pack2_bf16(float a, float b) {
  return ((unsigned int)(unsigned short)f32_to_bf16(b) << 16) |
         (unsigned int)(unsigned short)f32_to_bf16(a);
}

__device__ __forceinline__ void raw_barrier() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
}

template <int QPG>
__global__ __launch_bounds__(NW * WAVE, 2) void flash_prefill_glds_kernel(
    const short* __restrict__ q, const short* __restrict__ k_cache,
    const short* __restrict__ v_cache,
    const int32_t* __restrict__ block_tables,
    const int32_t* __restrict__ seq_meta, const int32_t* __restrict__ tiles,
    short* __restrict__ out, int kvh, int bs, int max_blocks, float scale) {
  const int kh = blockIdx.y;
  const int seq = tiles[2 * blockIdx.x];
  const int vrow0 = tiles[2 * blockIdx.x + 1];
  const int seq_start = seq_meta[3 * seq];
  const int chunk = seq_meta[3 * seq + 1];
  const int prior = seq_meta[3 * seq + 2];
  const int ctx = prior + chunk;
  const int n_q_heads = kvh * QPG;

  // ONE shared object (a second __shared__ symbol makes hipcc emit
  // vmcnt(0) before every ds_read — guide §5 trap (a))
  __shared__ __align__(16) char lds_raw[BT_CAP * 4 +
                                        NBUF * 2 * TILE_SHORTS * 2];
  int32_t* bt_lds = (int32_t*)lds_raw;
  auto kbuf = [&](int b) -> short* {
    return (short*)(lds_raw + BT_CAP * 4) + b * 2 * TILE_SHORTS;
  };
  auto vbuf = [&](int b) -> short* { return kbuf(b) + TILE_SHORTS; };

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const int qcol = lane & 31;
  const int hi = lane >> 5;

  const int vrow = vrow0 + wave * QROWS + qcol;
  const bool active = vrow < chunk * QPG;
  const int p = active ? vrow / QPG : -1;
  const int g = active ? vrow % QPG : 0;
  const int p_c = active ? p : chunk - 1;
  const int qh = kh * QPG + g;
  const int64_t q_row = (int64_t)(seq_start + p_c) * n_q_heads + qh;

  const int p_max_wg =
      min((min(vrow0 + NW * QROWS, chunk * QPG) - 1) / QPG, chunk - 1);
  const int p_max_w =
      min((min(vrow0 + (wave + 1) * QROWS, chunk * QPG) - 1) / QPG, chunk - 1);
  const int kv_end_wg = min(ctx, prior + p_max_wg + 1);
  const int kv_end_w = prior + p_max_w + 1;
  const int n_kv_tiles = (kv_end_wg + KVBLK - 1) / KVBLK;

  // ---- prologue: Q fragments + block table; every ordinary global load
  //      completes before the first glds issues ----
  bf16x8 q_frag[D / 16];
#pragma unroll
  for (int kk = 0; kk < D / 16; ++kk)
    q_frag[kk] = *(const bf16x8*)(q + q_row * D + kk * 16 + hi * 8);
  // stage enough entries to cover the final tile's clamped pad rows
  // (t = min(tile0+row, ctx-1) can reach up to kv_end_wg + KVBLK - 1)
  const int n_bt = min(max_blocks,
                       (min(kv_end_wg + KVBLK, ctx) + bs - 1) / bs);
  for (int i = tid; i < n_bt; i += NW * WAVE)
    bt_lds[i] = block_tables[(int64_t)seq * max_blocks + i];
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();  // publish bt_lds (no glds in flight yet)

  auto stage_tile = [&](int n) {
    const int tile0 = n * KVBLK;
    const int buf = n % NBUF;
#pragma unroll
    for (int j = 0; j < GLDS_PER_WAVE; ++j) {
      const int piece = (wave * GLDS_PER_WAVE + j) * 64 + lane;
      const int row = piece / 16, c16 = piece % 16;
      const int src16 = c16 ^ swz4(row);        // source-side swizzle
      const int t = min(tile0 + row, ctx - 1);  // clamp: no garbage bf16
      const int64_t base =
          (((int64_t)bt_lds[t / bs] * kvh + kh) * bs + t % bs) * D;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              k_cache + base + src16 * 8),
          (__attribute__((address_space(3))) unsigned int*)(
              (char*)kbuf(buf) + (wave * GLDS_PER_WAVE + j) * 1024),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              v_cache + base + src16 * 8),
          (__attribute__((address_space(3))) unsigned int*)(
              (char*)vbuf(buf) + (wave * GLDS_PER_WAVE + j) * 1024),
          16, 0, 0);
    }
  };

  float m_run = NEG, l_run = 0.f;
  f32x16 acc_o[D / 32];
#pragma unroll
  for (int dt = 0; dt < D / 32; ++dt) acc_o[dt] = (f32x16)(0.f);

  stage_tile(0);
  if (n_kv_tiles > 1) stage_tile(1);

  const int i16 = lane % 16;
  const int d_grp = 16 * ((lane & 31) >> 4);
  const int dim_col = d_grp + (i16 % 4) * 4;

  for (int n = 0; n < n_kv_tiles; ++n) {
    const int tile0 = n * KVBLK;
    const int buf = n % NBUF;
    // tile n has landed when at most the NEXT tile's DMA remains in
    // flight (counted per-wave wait; vmcnt retires in order)
    if (n + 1 < n_kv_tiles)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * GLDS_PER_WAVE)
                   : "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    raw_barrier();

    if (tile0 < kv_end_w) {
      const short* k_lds = kbuf(buf);
      const short* v_lds = vbuf(buf);
      f32x16 acc_s = (f32x16)(0.f);
#pragma unroll
      for (int kk = 0; kk < D / 16; ++kk) {
        const bf16x8 k_frag = *(const bf16x8*)(
            (const char*)k_lds + g_swz(qcol, (kk * 16 + hi * 8) * 2));
        acc_s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(k_frag, q_frag[kk],
                                                        acc_s, 0, 0, 0);
      }

      float s[16];
      float tmax = NEG;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv = tile0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const bool valid = active && kv <= prior + p;
        s[r] = valid ? acc_s[r] * scale : NEG;
        tmax = fmaxf(tmax, s[r]);
      }
      tmax = fmaxf(tmax, __shfl_xor(tmax, 32, WAVE));
      const float m_new = fmaxf(m_run, tmax);
      // MFMA reads operands from all 64 lanes regardless of EXEC: run the
      // PV block wave-uniformly, inactive lanes contribute zero P columns
      const bool has = m_new > NEG * 0.5f;
      if (__any(has)) {
        const float m_eff = has ? m_new : 0.f;
        const float alpha =
            (m_run <= NEG * 0.5f) ? 0.f : __expf(m_run - m_eff);
        float tsum = 0.f;
        float pr[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          pr[r] = (s[r] <= NEG * 0.5f) ? 0.f : __expf(s[r] - m_eff);
          tsum += pr[r];
        }
        tsum += __shfl_xor(tsum, 32, WAVE);
        l_run = l_run * alpha + tsum;
        if (has) m_run = m_new;
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
          for (int e = 0; e < 16; ++e) acc_o[dt][e] *= alpha;

        unsigned int w[8], x[8];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          w[i] = pack2_bf16(pr[2 * i], pr[2 * i + 1]);
          x[i] = __shfl_xor((int)w[i], 32, WAVE);
        }
        uint4v f0 = hi ? (uint4v){x[2], x[3], w[2], w[3]}
                       : (uint4v){w[0], w[1], x[0], x[1]};
        uint4v f1 = hi ? (uint4v){x[6], x[7], w[6], w[7]}
                       : (uint4v){w[4], w[5], x[4], x[5]};
        const bf16x8 p_frag0 = __builtin_bit_cast(bf16x8, f0);
        const bf16x8 p_frag1 = __builtin_bit_cast(bf16x8, f1);

#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt) {
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) {
            const int kv0 = ks * 16 + hi * 8 + i16 / 4;
            const int dc = (dt * 32 + dim_col) * 2;
            short4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                (__attribute__((address_space(3))) short4v*)(
                    (const char*)v_lds + kv0 * 256 +
                    (dc ^ (swz4(kv0) << 4))));
            const int kv1 = kv0 + 4;
            short4v hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                (__attribute__((address_space(3))) short4v*)(
                    (const char*)v_lds + kv1 * 256 +
                    (dc ^ (swz4(kv1) << 4))));
            short8 vfrag8 = {lo[0], lo[1], lo[2], lo[3],
                             hi4[0], hi4[1], hi4[2], hi4[3]};
            acc_o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                __builtin_bit_cast(bf16x8, vfrag8),
                ks == 0 ? p_frag0 : p_frag1, acc_o[dt], 0, 0, 0);
          }
        }
      }
    }
    if (n + 2 < n_kv_tiles) stage_tile(n + 2);
  }

  if (active && l_run > 0.f) {
    const float inv_l = 1.f / l_run;
    const int64_t out_row = ((int64_t)(seq_start + p) * n_q_heads + qh) * D;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt) {
#pragma unroll
      for (int rq = 0; rq < 4; ++rq) {
        const int dim = dt * 32 + 8 * rq + 4 * hi;
        unsigned int lo = pack2_bf16(acc_o[dt][4 * rq] * inv_l,
                                     acc_o[dt][4 * rq + 1] * inv_l);
        unsigned int hi2 = pack2_bf16(acc_o[dt][4 * rq + 2] * inv_l,
                                      acc_o[dt][4 * rq + 3] * inv_l);
        *(uint2*)(out + out_row + dim) = make_uint2(lo, hi2);
      }
    }
  }
}

}  // namespace

extern "C" {

hipError_t lds_flash_prefill_glds(
    const void* q, const void* k_cache, const void* v_cache,
    const int32_t* block_tables, const int32_t* seq_meta,
    const int32_t* tiles, void* out, int n_tiles, int n_q_heads, int kvh,
    int bs, int head_dim, int max_blocks, float scale, hipStream_t stream) {
  if (n_tiles == 0) return hipSuccess;
  if (head_dim != D || max_blocks > BT_CAP) return hipErrorInvalidValue;
  const int qpg = n_q_heads / kvh;
  dim3 grid(n_tiles, kvh), block(NW * WAVE);
#define LAUNCH_G(QPG)                                                        \
  hipLaunchKernelGGL((flash_prefill_glds_kernel<QPG>), grid, block, 0,       \
                     stream, (const short*)q, (const short*)k_cache,         \
                     (const short*)v_cache, block_tables, seq_meta, tiles,   \
                     (short*)out, kvh, bs, max_blocks, scale)
  switch (qpg) {
    case 1: LAUNCH_G(1); break;
    case 2: LAUNCH_G(2); break;
    case 4: LAUNCH_G(4); break;
    case 8: LAUNCH_G(8); break;
    default: return hipErrorInvalidValue;
  }
#undef LAUNCH_G
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
