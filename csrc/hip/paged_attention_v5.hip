// paged_attention_v5.hip — barrier-free-loop GQA decode attention:
// per-wave independent online softmax.
//
// Round-2 ladder: v1 (4-wave phased, 3 barriers/chunk) reaches 3.1-3.3
// TB/s vs the 6.5 TB/s its access pattern sustains (tools/kv_bw_probe.hip);
// v4 (producer/consumer waves) measured SLOWER — splitting the streams
// halved each stream's memory-level parallelism and the barrier coupled
// them (profiles/r02_notes.md). v5 removes the coupling a third way:
// every wave runs the WHOLE pipeline (K dots -> softmax -> V accumulate)
// over its own strided 64-token runs with PRIVATE (m, l, o) state, so
// there is NO workgroup barrier in the streaming loop at all — waves
// drift freely and the K and V streams of different waves interleave in
// the memory system. One final barrier merges the four per-wave partials
// exactly like flash-decoding partition merge:
//   m* = max_w m_w ; o = sum_w exp(m_w - m*) o_w ; l likewise.
// p-values cross lanes (lane=token -> lane=dim) through a per-wave LDS
// slice; LDS ops within one wave complete in order, so only a
// scheduling-fence (wave_barrier) is needed, not s_barrier.
#include "hip_common.h"

namespace {

constexpr int D = 128;
constexpr int RUN = 64;       // tokens per per-wave run (one K row / lane)
constexpr int NW = 4;
constexpr float NEG = -1e30f;

template <int QPG, bool SPLIT, typename CT>
__global__ __launch_bounds__(NW * WAVE) void paged_attention_v5_kernel(
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
  __shared__ float p_lds[NW][QPG][RUN];   // per-wave p-value slice
  __shared__ float comb_o[NW][QPG][D];
  __shared__ float comb_ml[NW][QPG][2];

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;

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
  __syncthreads();

  const int32_t* bt = block_tables + (int64_t)b * max_blocks;

  float m_w[QPG], l_w[QPG], o_acc[QPG][2];
#pragma unroll
  for (int h = 0; h < QPG; ++h) {
    m_w[h] = NEG;
    l_w[h] = 0.f;
    o_acc[h][0] = o_acc[h][1] = 0.f;
  }

  // wave-strided runs: wave w owns tokens [t_begin + (k*NW+w)*RUN, +RUN)
  for (int run0 = t_begin + wave * RUN; run0 < t_end; run0 += NW * RUN) {
    const int n_t = min(RUN, t_end - run0);
    // ---- K dot: lane owns one token's K row ----
    float dot[QPG];
#pragma unroll
    for (int h = 0; h < QPG; ++h) dot[h] = NEG;
    if (lane < n_t) {
      const int t = run0 + lane;
      const CT* krow =
          k_cache + ((((int64_t)bt[t / bs]) * kvh + kh) * bs + t % bs) * D;
      float acc[QPG];
#pragma unroll
      for (int h = 0; h < QPG; ++h) acc[h] = 0.f;
#pragma unroll 8
      for (int c = 0; c < D / 8; ++c) {
        float kf[8];
        load_kv8(krow + c * 8, kf);
#pragma unroll
        for (int h = 0; h < QPG; ++h) {
          const float4v* q4 = (const float4v*)&q_lds[h][c * 8];
          const float4v qa = q4[0], qb = q4[1];
          acc[h] += qa[0] * kf[0] + qa[1] * kf[1] + qa[2] * kf[2] +
                    qa[3] * kf[3] + qb[0] * kf[4] + qb[1] * kf[5] +
                    qb[2] * kf[6] + qb[3] * kf[7];
        }
      }
#pragma unroll
      for (int h = 0; h < QPG; ++h) dot[h] = acc[h];
    }
    // ---- per-wave online softmax over the 64 lane logits ----
#pragma unroll
    for (int h = 0; h < QPG; ++h) {
      const float lmax = wave_reduce_max(dot[h]);
      const float m_new = fmaxf(m_w[h], lmax);
      const float p = (dot[h] <= NEG) ? 0.f : __expf(dot[h] - m_new);
      const float lsum = wave_reduce_sum(p);
      const float alpha = (m_w[h] <= NEG) ? 0.f : __expf(m_w[h] - m_new);
      l_w[h] = l_w[h] * alpha + lsum;
      m_w[h] = m_new;
      p_lds[wave][h][lane] = p;
      o_acc[h][0] *= alpha;
      o_acc[h][1] *= alpha;
    }
    __builtin_amdgcn_wave_barrier();   // order p_lds writes before reads
    // ---- V accumulate: lane owns a dim pair, 8 rows in flight ----
    {
      int i = 0;
      for (; i + 8 <= n_t; i += 8) {
        const CT* vrows[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int t = run0 + i + j;
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
            const float p = p_lds[wave][h][i + j];
            o_acc[h][0] += p * vv[j][0];
            o_acc[h][1] += p * vv[j][1];
          }
        }
      }
      for (; i < n_t; ++i) {
        const int t = run0 + i;
        float v0, v1;
        load_kv2(v_cache + ((((int64_t)bt[t / bs]) * kvh + kh) * bs +
                            t % bs) * D + lane * 2, v0, v1);
#pragma unroll
        for (int h = 0; h < QPG; ++h) {
          const float p = p_lds[wave][h][i];
          o_acc[h][0] += p * v0;
          o_acc[h][1] += p * v1;
        }
      }
    }
    __builtin_amdgcn_wave_barrier();   // p_lds reuse next run
  }

  // ---- merge the four per-wave partials (flash-decoding style) ----
#pragma unroll
  for (int h = 0; h < QPG; ++h) {
    comb_o[wave][h][2 * lane] = o_acc[h][0];
    comb_o[wave][h][2 * lane + 1] = o_acc[h][1];
    if (lane == 0) {
      comb_ml[wave][h][0] = m_w[h];
      comb_ml[wave][h][1] = l_w[h];
    }
  }
  __syncthreads();
  if (SPLIT) {
    float* po = part_o + ((((int64_t)b * kvh + kh) * np + part) * QPG) * D;
    for (int i = tid; i < QPG * D; i += NW * WAVE) {
      const int h = i / D, d = i % D;
      float mg = NEG;
#pragma unroll
      for (int w = 0; w < NW; ++w) mg = fmaxf(mg, comb_ml[w][h][0]);
      float s = 0.f;
#pragma unroll
      for (int w = 0; w < NW; ++w) {
        const float ml0 = comb_ml[w][h][0];
        s += (ml0 <= NEG) ? 0.f : __expf(ml0 - mg) * comb_o[w][h][d];
      }
      po[h * D + d] = s;                  // unnormalized at max mg
    }
    if (tid < QPG) {
      float mg = NEG, lg = 0.f;
#pragma unroll
      for (int w = 0; w < NW; ++w) mg = fmaxf(mg, comb_ml[w][tid][0]);
#pragma unroll
      for (int w = 0; w < NW; ++w) {
        const float ml0 = comb_ml[w][tid][0];
        lg += (ml0 <= NEG) ? 0.f : __expf(ml0 - mg) * comb_ml[w][tid][1];
      }
      float* ml = part_ml + ((((int64_t)b * kvh + kh) * np + part) * QPG +
                             tid) * 2;
      ml[0] = mg;
      ml[1] = lg;
    }
  } else {
    for (int i = tid; i < QPG * D; i += NW * WAVE) {
      const int h = i / D, d = i % D;
      float mg = NEG;
#pragma unroll
      for (int w = 0; w < NW; ++w) mg = fmaxf(mg, comb_ml[w][h][0]);
      float s = 0.f, lg = 0.f;
#pragma unroll
      for (int w = 0; w < NW; ++w) {
        const float ml0 = comb_ml[w][h][0];
        if (ml0 <= NEG) continue;
        const float sc = __expf(ml0 - mg);
        s += sc * comb_o[w][h][d];
        lg += sc * comb_ml[w][h][1];
      }
      out[((int64_t)b * n_q_heads + qh0 + h) * D + d] =
          f32_to_bf16(lg > 0.f ? s / lg : 0.f);
    }
  }
}

// combine kernel for the SPLIT path (partition merge, same math)
template <int QPG>
__global__ __launch_bounds__(256) void v5_combine_kernel(
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

hipError_t lds_paged_attention_v5(
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
  hipLaunchKernelGGL((paged_attention_v5_kernel<QPG, SPLIT, CT>), grid,       \
                     block, 0, stream, (const short*)q, (const CT*)k_cache,   \
                     (const CT*)v_cache, block_tables, seq_lens,              \
                     (short*)out, part_o, part_ml, kvh, bs, max_blocks,       \
                     part_tokens, scale)
#define LAUNCH(QPG)                                                           \
  do {                                                                        \
    if (split) {                                                              \
      if (kv_fp8) LAUNCH_CT(QPG, true, unsigned char);                        \
      else LAUNCH_CT(QPG, true, short);                                       \
      hipLaunchKernelGGL((v5_combine_kernel<QPG>), cgrid, cblock, 0, stream,  \
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
