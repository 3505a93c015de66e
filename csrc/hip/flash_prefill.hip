// flash_prefill.hip — fused causal varlen GQA prefill attention over the
// paged KV pool, MFMA-tiled for gfx950 (CDNA4).
//
// Replaces the composed einsum+softmax prefill path (materialized S matrix,
// ~6 HBM passes) with a flash-style kernel: per Q-tile, iterate KV tiles
// computing QK^T -> online softmax -> P.V entirely on matrix cores
// (v_mfma_f32_32x32x16_bf16), S never leaving registers (guide Appendix B
// "Fused attention prefill").
//
// Layout choice ("swapped QK^T", guide §Appendix B): compute S^T = K.Q^T so
// each lane owns ONE query row (col = lane&31) across all its accumulator
// registers — the online-softmax row reduction is in-lane fmax/adds plus a
// single cross-half shuffle, and the O-rescale is a per-lane scalar.
//
// Work decomposition:
//   grid.x = row tiles (128 "virtual rows" = (position, q-head-in-group)
//            pairs, v = p*QPG + g), grid.y = kv_head
//   workgroup = 4 waves; wave w owns v-rows [vrow0+32w, vrow0+32w+32)
//   KV loop: 32-token tiles staged in LDS (K XOR-swizzled for
//            conflict-free ds_read_b128; V transposed with padded stride)
//
// GQA: q-heads of one group share the kv_head's K/V tiles; causal masking
// is per-lane compares (kv <= prior + p), no divergence.
#include "hip_common.h"

namespace {

constexpr int D = 128;       // head_dim
constexpr int KVBLK = 64;    // tokens per KV tile (2 mfma sub-tiles)
constexpr int NW = 4;        // waves per workgroup
constexpr int QROWS = 32;    // q virtual-rows per wave
constexpr float NEG = -1e30f;

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;

// K tile LDS: [32 tokens][256 B], 16-B chunks XOR-swizzled by (row&15)<<4
// so a 16-lane ds_read_b128 group (distinct row&15) is conflict-free.
__device__ __forceinline__ int k_swz(int row, int byte_off) {
  return row * 256 + (byte_off ^ ((row & 15) << 4));
}

// V tile is stored ROW-MAJOR with the same XOR swizzle as K; the PV
// A-operand (V^T fragments) is gathered by the gfx950 hardware transpose
// read v_ds_read_b64_tr_b16: within each fixed 16-lane group, lane i's
// element j comes from the group's combined 64-short window at
// [i + 16*j] — pointing lane i at &V[kv0 + i/4][d0 + (i%4)*4] delivers
// exactly A[dim=d0+i][kv=kv0+j] (layout verified empirically,
// tools/tr16_probe.hip). This removes the per-element b16
// scatter-transpose staging that dominated SQ_LDS_BANK_CONFLICT
// (profiles/r01_pmc_counters.md).
typedef __attribute__((ext_vector_type(4))) short short4v;

__device__ __forceinline__ unsigned int pack_bf16(float a, float b) {
  return ((unsigned int)(unsigned short)f32_to_bf16(b) << 16) |
         (unsigned int)(unsigned short)f32_to_bf16(a);
}

template <int QPG, typename CT>
__global__ __launch_bounds__(NW * WAVE, 2) void flash_prefill_kernel(
    const short* __restrict__ q,        // [T, QH, D]
    const CT* __restrict__ k_cache,     // [NB, KVH, BS, D] bf16|fp8
    const CT* __restrict__ v_cache,     // [NB, KVH, BS, D] bf16|fp8
    const int32_t* __restrict__ block_tables,  // [S, max_blocks]
    const int32_t* __restrict__ seq_meta,      // [S, 3] start, chunk, prior
    const int32_t* __restrict__ tiles,         // [n_tiles, 2] seq, vrow0
    short* __restrict__ out,                   // [T, QH, D]
    int kvh, int bs, int max_blocks, float scale) {
  const int kh = blockIdx.y;
  const int seq = tiles[2 * blockIdx.x];
  const int vrow0 = tiles[2 * blockIdx.x + 1];
  const int seq_start = seq_meta[3 * seq];
  const int chunk = seq_meta[3 * seq + 1];
  const int prior = seq_meta[3 * seq + 2];
  const int ctx = prior + chunk;
  const int n_q_heads = kvh * QPG;
  const int32_t* bt = block_tables + (int64_t)seq * max_blocks;

  __shared__ short k_lds[KVBLK * D];          // swizzled, byte-addressed
  __shared__ short v_lds[KVBLK * D];          // row-major, same swizzle
  static_assert(KVBLK * 16 % (NW * WAVE) == 0, "staging divides evenly");

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const int qcol = lane & 31;
  const int hi = lane >> 5;

  // this lane's query row
  const int vrow = vrow0 + wave * QROWS + qcol;
  const bool active = vrow < chunk * QPG;
  const int p = active ? vrow / QPG : -1;            // position in chunk
  const int g = active ? vrow % QPG : 0;
  const int p_c = active ? p : chunk - 1;            // clamped for loads
  const int qh = kh * QPG + g;
  const int64_t q_row = (int64_t)(seq_start + p_c) * n_q_heads + qh;

  // Q^T B-fragments: q_frag[kk] covers head dims [16kk + 8hi, 16kk + 8hi+8)
  bf16x8 q_frag[D / 16];
#pragma unroll
  for (int kk = 0; kk < D / 16; ++kk)
    q_frag[kk] = *(const bf16x8*)(q + q_row * D + kk * 16 + hi * 8);

  // online-softmax state (per lane = per q-row) and O^T accumulators
  float m_run = NEG, l_run = 0.f;
  f32x16 acc_o[D / 32];
#pragma unroll
  for (int dt = 0; dt < D / 32; ++dt) acc_o[dt] = (f32x16)(0.f);

  // causal KV extents
  const int p_max_wg =
      min((min(vrow0 + NW * QROWS, chunk * QPG) - 1) / QPG, chunk - 1);
  const int p_max_w =
      min((min(vrow0 + (wave + 1) * QROWS, chunk * QPG) - 1) / QPG, chunk - 1);
  const int kv_end_wg = min(ctx, prior + p_max_wg + 1);
  const int kv_end_w = prior + p_max_w + 1;

  for (int tile0 = 0; tile0 < kv_end_wg; tile0 += KVBLK) {
    // ---- stage K (swizzled) and V into LDS ----
    __syncthreads();  // previous tile's reads complete before overwrite
#pragma unroll
    for (int rep = 0; rep < KVBLK * 16 / (NW * WAVE); ++rep) {
      const int ch = tid + rep * (NW * WAVE);
      const int row = ch / 16, c16 = ch % 16;
      const int t = tile0 + row;
      short8 piece = {0, 0, 0, 0, 0, 0, 0, 0};
      short8 vpiece = {0, 0, 0, 0, 0, 0, 0, 0};
      if (t < ctx) {
        const int64_t base =
            (((int64_t)bt[t / bs] * kvh + kh) * bs + t % bs) * D;
        piece = load_kv8_bf16(k_cache + base + c16 * 8);
        vpiece = load_kv8_bf16(v_cache + base + c16 * 8);
      }
      *(short8*)((char*)k_lds + k_swz(row, c16 * 16)) = piece;
      *(short8*)((char*)v_lds + k_swz(row, c16 * 16)) = vpiece;
    }
    __syncthreads();

    if (tile0 >= kv_end_w) continue;  // beyond this wave's causal horizon

#pragma unroll 1
    for (int half = 0; half < 2; ++half) {
    const int sub0 = tile0 + 32 * half;
    if (sub0 >= kv_end_w) break;
    // ---- S^T = K . Q^T : acc rows = kv tokens, cols = q rows ----
    f32x16 acc_s = (f32x16)(0.f);
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      const bf16x8 k_frag = *(const bf16x8*)(
          (const char*)k_lds + k_swz(32 * half + qcol,
                                     (kk * 16 + hi * 8) * 2));
      acc_s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(k_frag, q_frag[kk],
                                                      acc_s, 0, 0, 0);
    }

    // ---- per-lane online softmax over this sub-tile's 32 kv tokens ----
    float s[16];
    float tmax = NEG;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kv = sub0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      const bool valid = active && kv <= prior + p;
      s[r] = valid ? acc_s[r] * scale : NEG;
      tmax = fmaxf(tmax, s[r]);
    }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, WAVE));
    const float m_new = fmaxf(m_run, tmax);
    // MFMA reads operand registers from ALL 64 lanes regardless of EXEC,
    // so the PV block below must run wave-uniformly: lanes with no valid
    // element yet (m_new still NEG) contribute zero P columns instead of
    // branching around the matrix op (a per-lane `if` here corrupts the
    // shared A/B operands with uninitialized registers — measured).
    const bool has = m_new > NEG * 0.5f;
    if (__any(has)) {
      const float m_eff = has ? m_new : 0.f;
      const float alpha = (m_run <= NEG * 0.5f) ? 0.f : __expf(m_run - m_eff);
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

      // ---- pack P^T into B fragments (cross-half exchange) ----
      unsigned int w[8], x[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        w[i] = pack_bf16(pr[2 * i], pr[2 * i + 1]);
        x[i] = __shfl_xor((int)w[i], 32, WAVE);
      }
      uint4v f0 = hi ? (uint4v){x[2], x[3], w[2], w[3]}
                     : (uint4v){w[0], w[1], x[0], x[1]};
      uint4v f1 = hi ? (uint4v){x[6], x[7], w[6], w[7]}
                     : (uint4v){w[4], w[5], x[4], x[5]};
      const bf16x8 p_frag0 = __builtin_bit_cast(bf16x8, f0);
      const bf16x8 p_frag1 = __builtin_bit_cast(bf16x8, f1);

      // ---- O^T += V^T . P^T (A-operand via hardware transpose reads) ----
      const int i16 = lane % 16;
      const int d_grp = 16 * ((lane & 31) >> 4);   // group's dim base
      const int dim_col = d_grp + (i16 % 4) * 4;
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt) {
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const int kv0 = 32 * half + ks * 16 + hi * 8 + i16 / 4;
          const int dc = (dt * 32 + dim_col) * 2;
          short4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (__attribute__((address_space(3))) short4v*)(
                  (char*)v_lds + kv0 * 256 + (dc ^ ((kv0 & 15) << 4))));
          const int kv1 = kv0 + 4;
          short4v hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (__attribute__((address_space(3))) short4v*)(
                  (char*)v_lds + kv1 * 256 + (dc ^ ((kv1 & 15) << 4))));
          short8 vfrag8 = {lo[0], lo[1], lo[2], lo[3],
                           hi4[0], hi4[1], hi4[2], hi4[3]};
          acc_o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              __builtin_bit_cast(bf16x8, vfrag8),
              ks == 0 ? p_frag0 : p_frag1, acc_o[dt], 0, 0, 0);
        }
      }
    }
    }  // half
  }

  // ---- epilogue: out[seq_start+p][qh][dim] = acc_o / l ----
  if (active && l_run > 0.f) {
    const float inv_l = 1.f / l_run;
    const int64_t out_row = ((int64_t)(seq_start + p) * n_q_heads + qh) * D;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt) {
#pragma unroll
      for (int rq = 0; rq < 4; ++rq) {   // reg quads: dims 4-contiguous
        const int dim = dt * 32 + 8 * rq + 4 * hi;
        unsigned int lo = pack_bf16(acc_o[dt][4 * rq] * inv_l,
                                    acc_o[dt][4 * rq + 1] * inv_l);
        unsigned int hi2 = pack_bf16(acc_o[dt][4 * rq + 2] * inv_l,
                                     acc_o[dt][4 * rq + 3] * inv_l);
        *(uint2*)(out + out_row + dim) = make_uint2(lo, hi2);
      }
    }
  }
}

}  // namespace

extern "C" {

hipError_t lds_flash_prefill(const void* q, const void* k_cache,
                             const void* v_cache,
                             const int32_t* block_tables,
                             const int32_t* seq_meta, const int32_t* tiles,
                             void* out, int n_tiles, int n_q_heads, int kvh,
                             int bs, int head_dim, int max_blocks, int kv_fp8,
                             float scale, hipStream_t stream) {
  if (n_tiles == 0) return hipSuccess;
  if (head_dim != D) return hipErrorInvalidValue;
  const int qpg = n_q_heads / kvh;
  dim3 grid(n_tiles, kvh), block(NW * WAVE);
#define LAUNCH_CT(QPG, CT)                                                    \
  hipLaunchKernelGGL((flash_prefill_kernel<QPG, CT>), grid, block, 0, stream, \
                     (const short*)q, (const CT*)k_cache,                     \
                     (const CT*)v_cache, block_tables, seq_meta, tiles,       \
                     (short*)out, kvh, bs, max_blocks, scale)
#define LAUNCH(QPG)                                                           \
  do {                                                                        \
    if (kv_fp8) LAUNCH_CT(QPG, unsigned char);                                \
    else LAUNCH_CT(QPG, short);                                               \
  } while (0)
  switch (qpg) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 4: LAUNCH(4); break;
    case 8: LAUNCH(8); break;
    default: return hipErrorInvalidValue;
  }
#undef LAUNCH
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
