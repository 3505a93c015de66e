// paged_attention_v4.hip — producer/consumer pipelined GQA decode attention.
//
// Round-1 measurement (profiles/r01_summary.md): the 4-wave phased kernel
// (paged_attention.hip) reaches 3.1-3.3 TB/s while its ACCESS PATTERN alone
// sustains 6.5 TB/s (tools/kv_bw_probe.hip) — the gap is HBM duty cycle
// lost to phase serialization: K streams in phase A, nothing streams during
// softmax B, V streams in phase C, with three barriers per 256-token chunk.
// The barrier-free one-wave-per-unit v3 measured SLOWER (occupancy/latency
// trade) — this is the cross-chunk software-pipelining shape the round-1
// notes called for instead:
//
//   waves 0-1 (producers): K-row dots + online softmax for chunk n+1
//   waves 2-3 (consumers): V accumulation for chunk n
//   logits / alpha double-buffered in LDS; ONE barrier per chunk.
//
// Both HBM streams (K for n+1, V for n) are in flight between barriers, so
// the memory system never idles during softmax. Same math as v1: online
// softmax with per-chunk rescale; fp32 accumulation throughout.
#include "hip_common.h"

namespace {

constexpr int D = 128;
constexpr int CHUNK = 256;
constexpr int NW = 4;
constexpr float NEG = -1e30f;

template <int QPG, bool SPLIT, typename CT>
__global__ __launch_bounds__(NW * WAVE) void paged_attention_v4_kernel(
    const short* __restrict__ q,        // [B, QH, D]
    const CT* __restrict__ k_cache,     // [NB, KVH, BS, D]
    const CT* __restrict__ v_cache,
    const int32_t* __restrict__ block_tables,  // [B, max_blocks]
    const int32_t* __restrict__ seq_lens,      // [B]
    short* __restrict__ out,                   // [B, QH, D]
    float* __restrict__ part_o,   // [B, KVH, NP, QPG, D] (SPLIT)
    float* __restrict__ part_ml,  // [B, KVH, NP, QPG, 2] (SPLIT)
    int kvh, int bs, int max_blocks, int part_tokens, float scale) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int part = SPLIT ? blockIdx.z : 0;
  const int np = SPLIT ? gridDim.z : 1;
  const int qh0 = kh * QPG;
  const int n_q_heads = kvh * QPG;
  const int seq_len = seq_lens[b];
  const int t_begin = SPLIT ? part * part_tokens : 0;
  const int t_end = SPLIT ? min(seq_len, t_begin + part_tokens) : seq_len;

  __shared__ float q_lds[QPG][D];
  __shared__ float logits[2][QPG][CHUNK];
  __shared__ float m_sh[QPG], l_sh[QPG];   // producer-owned running stats
  __shared__ float alpha_sh[2][QPG];       // per-buffer rescale for consumers
  __shared__ float comb[2][QPG][D];        // consumer-wave combine

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const bool producer = wave < 2;
  const int pw = wave;        // producer wave id (0,1)
  const int cw = wave - 2;    // consumer wave id (0,1)

  if (SPLIT && t_begin >= seq_len) {
    if (tid < QPG) {
      float* ml = part_ml + ((((int64_t)b * kvh + kh) * np + part) * QPG +
                             tid) * 2;
      ml[0] = NEG;
      ml[1] = 0.f;
    }
    return;
  }

  for (int i = tid; i < QPG * D; i += NW * WAVE) {
    int h = i / D, d = i % D;
    q_lds[h][d] =
        bf16_to_f32(q[((int64_t)b * n_q_heads + qh0 + h) * D + d]) * scale;
  }
  if (tid < QPG) {
    m_sh[tid] = NEG;
    l_sh[tid] = 0.f;
  }
  __syncthreads();

  const int32_t* bt = block_tables + (int64_t)b * max_blocks;
  const int nc = (t_end - t_begin + CHUNK - 1) / CHUNK;

  // ---------------- pipeline protocol ----------------
  // Two stages (= two workgroup barriers) per chunk:
  //   stage A: producers write raw logits(i);   consumers V(i-1) half 0
  //   stage B: producers softmax(i) in place;   consumers V(i-1) half 1
  // The barrier after stage A is what makes BOTH producer waves' raw
  // logits visible for the full-chunk softmax; consumers use it to split
  // their V work so the V stream also covers the (HBM-idle) softmax stage.
  // V(i-1)'s p-values were finalized by stage B of the previous chunk.

  float o_acc[QPG][2];
#pragma unroll
  for (int h = 0; h < QPG; ++h) o_acc[h][0] = o_acc[h][1] = 0.f;

  // consumer V state: each consumer wave owns 128 tokens of the chunk,
  // split into two 64-token halves (one per stage)
  auto consume_half = [&](int ci, int buf, int half) {
    const int chunk0 = t_begin + ci * CHUNK;
    const int n_t = min(CHUNK, t_end - chunk0);
    const int t_base = cw * 2 * WAVE + half * WAVE;
    const int t_cnt = min(WAVE, n_t - t_base);
    if (half == 0) {
#pragma unroll
      for (int h = 0; h < QPG; ++h) {
        const float a = alpha_sh[buf][h];
        o_acc[h][0] *= a;
        o_acc[h][1] *= a;
      }
    }
    int i = 0;
    for (; i + 8 <= t_cnt; i += 8) {
      const CT* vrows[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int t = chunk0 + t_base + i + j;
        vrows[j] = v_cache + ((((int64_t)bt[t / bs]) * kvh + kh) * bs +
                              t % bs) * D + lane * 2;
      }
      float vv[8][2];
#pragma unroll
      for (int j = 0; j < 8; ++j) load_kv2(vrows[j], vv[j][0], vv[j][1]);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
#pragma unroll
        for (int h = 0; h < QPG; ++h) {
          const float p = logits[buf][h][t_base + i + j];
          o_acc[h][0] += p * vv[j][0];
          o_acc[h][1] += p * vv[j][1];
        }
      }
    }
    for (; i < t_cnt; ++i) {
      const int t = chunk0 + t_base + i;
      float v0, v1;
      load_kv2(v_cache + ((((int64_t)bt[t / bs]) * kvh + kh) * bs +
                          t % bs) * D + lane * 2, v0, v1);
#pragma unroll
      for (int h = 0; h < QPG; ++h) {
        const float p = logits[buf][h][t_base + i];
        o_acc[h][0] += p * v0;
        o_acc[h][1] += p * v1;
      }
    }
  };

  auto produce_raw = [&](int ci, int buf) {
    const int chunk0 = t_begin + ci * CHUNK;
    const int n_t = min(CHUNK, t_end - chunk0);
    const int ta = pw * WAVE + lane;
    const int tb = ta + 2 * WAVE;
    float dota[QPG], dotb[QPG];
#pragma unroll
    for (int h = 0; h < QPG; ++h) dota[h] = dotb[h] = 0.f;
    const bool va = ta < n_t, vb = tb < n_t;
    const CT* ka = nullptr;
    const CT* kb = nullptr;
    if (va) {
      const int t = chunk0 + ta;
      ka = k_cache + ((((int64_t)bt[t / bs]) * kvh + kh) * bs + t % bs) * D;
    }
    if (vb) {
      const int t = chunk0 + tb;
      kb = k_cache + ((((int64_t)bt[t / bs]) * kvh + kh) * bs + t % bs) * D;
    }
#pragma unroll 4
    for (int c = 0; c < D / 8; ++c) {
      float fa[8], fb[8];
      if (va) load_kv8(ka + c * 8, fa);
      if (vb) load_kv8(kb + c * 8, fb);
#pragma unroll
      for (int h = 0; h < QPG; ++h) {
        const float4v* q4 = (const float4v*)&q_lds[h][c * 8];
        const float4v qa = q4[0], qb = q4[1];
        if (va)
          dota[h] += qa[0] * fa[0] + qa[1] * fa[1] + qa[2] * fa[2] +
                     qa[3] * fa[3] + qb[0] * fa[4] + qb[1] * fa[5] +
                     qb[2] * fa[6] + qb[3] * fa[7];
        if (vb)
          dotb[h] += qa[0] * fb[0] + qa[1] * fb[1] + qa[2] * fb[2] +
                     qa[3] * fb[3] + qb[0] * fb[4] + qb[1] * fb[5] +
                     qb[2] * fb[6] + qb[3] * fb[7];
      }
    }
#pragma unroll
    for (int h = 0; h < QPG; ++h) {
      logits[buf][h][ta] = va ? dota[h] : NEG;
      logits[buf][h][tb] = vb ? dotb[h] : NEG;
    }
  };

  auto produce_softmax = [&](int ci, int buf) {
    // producer wave pw handles heads pw, pw+2, ... over the FULL chunk
    for (int h = pw; h < QPG; h += 2) {
      float lmax = NEG;
#pragma unroll
      for (int i = 0; i < CHUNK / WAVE; ++i)
        lmax = fmaxf(lmax, logits[buf][h][i * WAVE + lane]);
      lmax = wave_reduce_max(lmax);
      const float m_old = m_sh[h];
      const float m_new = fmaxf(m_old, lmax);
      float lsum = 0.f;
#pragma unroll
      for (int i = 0; i < CHUNK / WAVE; ++i) {
        const int idx = i * WAVE + lane;
        const float lg = logits[buf][h][idx];
        const float p = (lg <= NEG) ? 0.f : __expf(lg - m_new);
        logits[buf][h][idx] = p;
        lsum += p;
      }
      lsum = wave_reduce_sum(lsum);
      if (lane == 0) {
        const float alpha = (m_old <= NEG) ? 0.f : __expf(m_old - m_new);
        alpha_sh[buf][h] = alpha;
        l_sh[h] = l_sh[h] * alpha + lsum;
        m_sh[h] = m_new;
      }
    }
  };

  // pipeline: stage A of chunk i overlaps consumer half 1 of chunk i-1;
  // stage B (softmax i) overlaps consumer half 0 of... the halves lag one
  // stage behind the producer stages:
  //   barrier k   (producers: raw(i))      (consumers: V(i-1) half 1)
  //   barrier k+1 (producers: softmax(i))  (consumers: idle->V(i) half 0
  //                                         needs softmax(i) done... )
  // Consumers can only start chunk i's half 0 AFTER softmax(i) — so the
  // consumer schedule is: half0(i-1) during raw(i), half1(i-1) during
  // softmax(i). V(i-1) logits were finalized before raw(i) started.
  if (nc > 0) {
    if (producer) produce_raw(0, 0);
    __syncthreads();
    if (producer) produce_softmax(0, 0);
    __syncthreads();
    for (int i = 1; i <= nc; ++i) {
      const int buf = (i - 1) & 1;
      if (producer) {
        if (i < nc) produce_raw(i, buf ^ 1);
      } else {
        consume_half(i - 1, buf, 0);
      }
      __syncthreads();
      if (producer) {
        if (i < nc) produce_softmax(i, buf ^ 1);
      } else {
        consume_half(i - 1, buf, 1);
      }
      __syncthreads();
    }
  }

  // ---- combine the two consumer waves ----
  if (!producer) {
#pragma unroll
    for (int h = 0; h < QPG; ++h) {
      comb[cw][h][2 * lane] = o_acc[h][0];
      comb[cw][h][2 * lane + 1] = o_acc[h][1];
    }
  }
  __syncthreads();
  if (SPLIT) {
    float* po = part_o + ((((int64_t)b * kvh + kh) * np + part) * QPG) * D;
    for (int i = tid; i < QPG * D; i += NW * WAVE) {
      const int h = i / D, d = i % D;
      po[h * D + d] = comb[0][h][d] + comb[1][h][d];
    }
    if (tid < QPG) {
      float* ml = part_ml + ((((int64_t)b * kvh + kh) * np + part) * QPG +
                             tid) * 2;
      ml[0] = m_sh[tid];
      ml[1] = l_sh[tid];
    }
  } else {
    for (int i = tid; i < QPG * D; i += NW * WAVE) {
      const int h = i / D, d = i % D;
      const float l = l_sh[h];
      out[((int64_t)b * n_q_heads + qh0 + h) * D + d] =
          f32_to_bf16(l > 0.f ? (comb[0][h][d] + comb[1][h][d]) / l : 0.f);
    }
  }
}

// combine kernel for the SPLIT path (same math as paged_attention.hip)
template <int QPG>
__global__ __launch_bounds__(256) void v4_combine_kernel(
    const float* __restrict__ part_o, const float* __restrict__ part_ml,
    short* __restrict__ out, int kvh, int np) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int n_q_heads = kvh * QPG;
  __shared__ float m_g[QPG], scale_p[64][QPG];
  const int tid = threadIdx.x;
  if (tid < QPG) {
    float m = NEG;
    for (int p = 0; p < np; ++p)
      m = fmaxf(m, part_ml[((((int64_t)b * kvh + kh) * np + p) * QPG + tid)
                           * 2]);
    m_g[tid] = m;
  }
  __syncthreads();
  for (int i = tid; i < QPG * np; i += 256) {
    const int p = i / QPG, h = i % QPG;
    const float* ml =
        part_ml + ((((int64_t)b * kvh + kh) * np + p) * QPG + h) * 2;
    scale_p[p][h] = (ml[1] > 0.f) ? __expf(ml[0] - m_g[h]) : 0.f;
  }
  __syncthreads();
  for (int i = tid; i < QPG * D; i += 256) {
    const int h = i / D, d = i % D;
    float o = 0.f, l = 0.f;
    for (int p = 0; p < np; ++p) {
      const float s = scale_p[p][h];
      if (s == 0.f) continue;
      o += s * part_o[((((int64_t)b * kvh + kh) * np + p) * QPG + h) * D + d];
      l += s * part_ml[((((int64_t)b * kvh + kh) * np + p) * QPG + h) * 2 + 1];
    }
    out[((int64_t)b * n_q_heads + kh * QPG + h) * D + d] =
        f32_to_bf16(l > 0.f ? o / l : 0.f);
  }
}

}  // namespace

extern "C" {

hipError_t lds_paged_attention_v4(
    const void* q, const void* k_cache, const void* v_cache,
    const int32_t* block_tables, const int32_t* seq_lens, void* out,
    float* part_o, float* part_ml, int n_seqs, int n_q_heads, int kvh, int bs,
    int head_dim, int max_blocks, int n_parts, int part_tokens, int kv_fp8,
    float scale, hipStream_t stream) {
  if (n_seqs == 0) return hipSuccess;
  if (head_dim != D || n_parts > 64) return hipErrorInvalidValue;
  const int qpg = n_q_heads / kvh;
  const bool split = n_parts > 1;
  dim3 grid(n_seqs, kvh, split ? n_parts : 1), block(NW * WAVE);
  dim3 cgrid(n_seqs, kvh), cblock(256);
#define LAUNCH_CT(QPG, SPLIT, CT)                                             \
  hipLaunchKernelGGL((paged_attention_v4_kernel<QPG, SPLIT, CT>), grid,       \
                     block, 0, stream, (const short*)q, (const CT*)k_cache,   \
                     (const CT*)v_cache, block_tables, seq_lens,              \
                     (short*)out, part_o, part_ml, kvh, bs, max_blocks,       \
                     part_tokens, scale)
#define LAUNCH(QPG)                                                           \
  do {                                                                        \
    if (split) {                                                              \
      if (kv_fp8) LAUNCH_CT(QPG, true, unsigned char);                        \
      else LAUNCH_CT(QPG, true, short);                                       \
      hipLaunchKernelGGL((v4_combine_kernel<QPG>), cgrid, cblock, 0, stream,  \
                         part_o, part_ml, (short*)out, kvh, n_parts);         \
    } else {                                                                  \
      if (kv_fp8) LAUNCH_CT(QPG, false, unsigned char);                       \
      else LAUNCH_CT(QPG, false, short);                                      \
    }                                                                         \
  } while (0)
  switch (qpg) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 4: LAUNCH(4); break;
    case 8: LAUNCH(8); break;
    default: return hipErrorInvalidValue;
  }
#undef LAUNCH
#undef LAUNCH_CT
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
