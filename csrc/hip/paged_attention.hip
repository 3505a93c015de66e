// paged_attention.hip — GQA decode attention over the paged KV pool (gfx950).
//
// The decode-step hot kernel of the per-GPU worker engine. Memory-bound: the
// whole point is streaming each sequence's K/V exactly once per step at
// near-HBM rate (guide Appendix B "Attention decode").
//
// Geometry: one workgroup (4 waves, 256 threads) per (sequence, kv_head,
// partition). Q-heads in the GQA group (QPG = n_q_heads / n_kv_heads, <= 8)
// are scored together so K rows are read once for the whole group.
//   Phase A  each wave owns 64 tokens of a 256-token chunk; a lane streams
//            its token's K row (short8 loads) against the group's Q vectors
//            staged in LDS (broadcast reads) -> logits in LDS.
//   Phase B  online-softmax update per q-head (wave-parallel reductions).
//   Phase C  V accumulation: lane owns a dim pair, wave owns its 64 tokens;
//            V rows stream fully coalesced (64 lanes x 4 B = one 256 B row).
//   Final    cross-wave combine via LDS, normalize, bf16 store.
//
// Flash-decoding sequence split (SPLIT=true): long sequences are cut into
// gridDim.z partitions so the launch has >> 256 workgroups (the chip needs
// ~2 WGs/CU just to be full; B*KVH alone is often < 512). Each partition
// writes an UNNORMALIZED partial (o_acc, m, l) and a tiny combine kernel
// merges partitions: m* = max m_p; o = sum o_p*exp(m_p-m*); l likewise.
#include "hip_common.h"

namespace {

constexpr int D = 128;        // head_dim (Llama-3 family)
constexpr int DEF_CHUNK = 256;  // default tokens per online-softmax chunk
constexpr int NW = 4;         // waves per workgroup
constexpr float NEG = -1e30f;

template <int QPG, bool SPLIT, typename CT, int CHUNK = DEF_CHUNK>
__global__ __launch_bounds__(NW * WAVE) void paged_attention_kernel(
    const short* __restrict__ q,        // [B, QH, D]
    const CT* __restrict__ k_cache,     // [NB, KVH, BS, D] bf16|fp8
    const CT* __restrict__ v_cache,     // [NB, KVH, BS, D] bf16|fp8
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
  __shared__ float logits[QPG][CHUNK];
  __shared__ float m_sh[QPG], l_sh[QPG], alpha_sh[QPG];
  __shared__ float comb[NW][QPG][D];

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;

  if (SPLIT && t_begin >= seq_len) {
    // empty partition: publish l=0 so the combiner skips it
    if (tid < QPG) {
      float* ml = part_ml + ((((int64_t)b * kvh + kh) * np + part) * QPG +
                             tid) * 2;
      ml[0] = NEG;
      ml[1] = 0.f;
    }
    return;
  }

  // stage scaled Q for the group into LDS
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

  float o_acc[QPG][2];
#pragma unroll
  for (int h = 0; h < QPG; ++h) o_acc[h][0] = o_acc[h][1] = 0.f;

  const int32_t* bt = block_tables + (int64_t)b * max_blocks;

  for (int chunk0 = t_begin; chunk0 < t_end; chunk0 += CHUNK) {
    const int n_t = min(CHUNK, t_end - chunk0);
    // ---- Phase A: logits[h][t_local] (CHUNK may exceed one row/lane:
    // each pass of the tb loop covers NW*WAVE tokens) ----
    for (int tb = 0; tb < CHUNK; tb += NW * WAVE) {
      const int t_local = tb + wave * WAVE + lane;
      float dot[QPG];
#pragma unroll
      for (int h = 0; h < QPG; ++h) dot[h] = 0.f;
      if (t_local < n_t) {
        const int t = chunk0 + t_local;
        const int64_t blk = bt[t / bs];
        const int row = t % bs;
        const CT* krow = k_cache + (((blk * kvh + kh) * bs) + row) * D;
#pragma unroll 8
        for (int c = 0; c < D / 8; ++c) {
          float kf[8];
          load_kv8(krow + c * 8, kf);
#pragma unroll
          for (int h = 0; h < QPG; ++h) {
            const float4v* q4 = (const float4v*)&q_lds[h][c * 8];
            float4v qa = q4[0], qb = q4[1];
            dot[h] += qa[0] * kf[0] + qa[1] * kf[1] + qa[2] * kf[2] +
                      qa[3] * kf[3] + qb[0] * kf[4] + qb[1] * kf[5] +
                      qb[2] * kf[6] + qb[3] * kf[7];
          }
        }
      }
#pragma unroll
      for (int h = 0; h < QPG; ++h)
        logits[h][t_local] = (t_local < n_t) ? dot[h] : NEG;
    }
    __syncthreads();

    // ---- Phase B: online softmax per head (wave w handles head w, w+NW..) ----
    for (int h = wave; h < QPG; h += NW) {
      float lmax = NEG;
#pragma unroll
      for (int i = 0; i < CHUNK / WAVE; ++i)
        lmax = fmaxf(lmax, logits[h][i * WAVE + lane]);
      lmax = wave_reduce_max(lmax);
      const float m_old = m_sh[h];
      const float m_new = fmaxf(m_old, lmax);
      float lsum = 0.f;
#pragma unroll
      for (int i = 0; i < CHUNK / WAVE; ++i) {
        const int idx = i * WAVE + lane;
        float p = (logits[h][idx] <= NEG) ? 0.f : __expf(logits[h][idx] - m_new);
        logits[h][idx] = p;
        lsum += p;
      }
      lsum = wave_reduce_sum(lsum);
      if (lane == 0) {
        const float alpha = (m_old <= NEG) ? 0.f : __expf(m_old - m_new);
        alpha_sh[h] = alpha;
        l_sh[h] = l_sh[h] * alpha + lsum;
        m_sh[h] = m_new;
      }
    }
    __syncthreads();

    // ---- Phase C: V accumulation (wave owns its 64 tokens, lane owns 2 dims) ----
#pragma unroll
    for (int h = 0; h < QPG; ++h) {
      const float a = alpha_sh[h];
      o_acc[h][0] *= a;
      o_acc[h][1] *= a;
    }
    {
      const int t_base = wave * (CHUNK / NW);
      const int t_cnt = min(CHUNK / NW, n_t - t_base);
      // batch the V row loads 8 deep so they overlap (each row is one
      // coalesced 256-B wave read; a load-use loop serializes on latency)
      int i = 0;
      for (; i + 8 <= t_cnt; i += 8) {
        const CT* vrows[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int t = chunk0 + t_base + i + j;
          const int64_t blk = bt[t / bs];
          vrows[j] = v_cache + (((blk * kvh + kh) * bs) + t % bs) * D +
                     lane * 2;
        }
        float vv[8][2];
#pragma unroll
        for (int j = 0; j < 8; ++j) load_kv2(vrows[j], vv[j][0], vv[j][1]);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
#pragma unroll
          for (int h = 0; h < QPG; ++h) {
            const float p = logits[h][t_base + i + j];
            o_acc[h][0] += p * vv[j][0];
            o_acc[h][1] += p * vv[j][1];
          }
        }
      }
      for (; i < t_cnt; ++i) {
        const int t = chunk0 + t_base + i;
        const int64_t blk = bt[t / bs];
        float v0, v1;
        load_kv2(v_cache + (((blk * kvh + kh) * bs) + t % bs) * D + lane * 2,
                 v0, v1);
#pragma unroll
        for (int h = 0; h < QPG; ++h) {
          const float p = logits[h][t_base + i];
          o_acc[h][0] += p * v0;
          o_acc[h][1] += p * v1;
        }
      }
    }
    __syncthreads();  // logits reused next chunk
  }

  // ---- combine partial o across waves ----
#pragma unroll
  for (int h = 0; h < QPG; ++h) {
    comb[wave][h][2 * lane] = o_acc[h][0];
    comb[wave][h][2 * lane + 1] = o_acc[h][1];
  }
  __syncthreads();
  if (SPLIT) {
    float* po = part_o + ((((int64_t)b * kvh + kh) * np + part) * QPG) * D;
    for (int i = tid; i < QPG * D; i += NW * WAVE) {
      const int h = i / D, d = i % D;
      float s = 0.f;
#pragma unroll
      for (int w = 0; w < NW; ++w) s += comb[w][h][d];
      po[h * D + d] = s;                     // unnormalized
    }
    if (tid < QPG) {
      float* ml = part_ml + ((((int64_t)b * kvh + kh) * np + part) * QPG +
                             tid) * 2;
      ml[0] = m_sh[tid];
      ml[1] = l_sh[tid];
    }
  } else {
    // 256 threads cover QPG*D outputs (QPG<=8 -> <=1024 values, loop)
    for (int i = tid; i < QPG * D; i += NW * WAVE) {
      const int h = i / D, d = i % D;
      float s = 0.f;
#pragma unroll
      for (int w = 0; w < NW; ++w) s += comb[w][h][d];
      const float l = l_sh[h];
      out[((int64_t)b * n_q_heads + qh0 + h) * D + d] =
          f32_to_bf16(l > 0.f ? s / l : 0.f);
    }
  }
}

// Merge the NP partitions of one (seq, kv_head): one workgroup per
// (seq, kv_head), each thread owns (head, dim) outputs striding QPG*D.
template <int QPG>
__global__ __launch_bounds__(256) void paged_attention_combine_kernel(
    const float* __restrict__ part_o,   // [B, KVH, NP, QPG, D]
    const float* __restrict__ part_ml,  // [B, KVH, NP, QPG, 2]
    short* __restrict__ out,            // [B, QH, D]
    int kvh, int np) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int n_q_heads = kvh * QPG;
  __shared__ float m_g[QPG], scale_p[64][QPG];  // np <= 64

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

hipError_t lds_paged_attention(const void* q, const void* k_cache,
                               const void* v_cache, const int32_t* block_tables,
                               const int32_t* seq_lens, void* out, int n_seqs,
                               int n_q_heads, int kvh, int bs, int head_dim,
                               int max_blocks, int kv_fp8, int chunk,
                               float scale, hipStream_t stream) {
  if (n_seqs == 0) return hipSuccess;
  if (head_dim != D) return hipErrorInvalidValue;
  if (chunk != 256 && chunk != 512) return hipErrorInvalidValue;
  const int qpg = n_q_heads / kvh;
  dim3 grid(n_seqs, kvh), block(NW * WAVE);
#define LAUNCH_CT(QPG, CT)                                                    \
  do {                                                                        \
    if (chunk == 512)                                                         \
      hipLaunchKernelGGL((paged_attention_kernel<QPG, false, CT, 512>),       \
                         grid, block, 0, stream, (const short*)q,             \
                         (const CT*)k_cache, (const CT*)v_cache,              \
                         block_tables, seq_lens, (short*)out, nullptr,        \
                         nullptr, kvh, bs, max_blocks, 0, scale);             \
    else                                                                      \
      hipLaunchKernelGGL((paged_attention_kernel<QPG, false, CT, 256>),       \
                         grid, block, 0, stream, (const short*)q,             \
                         (const CT*)k_cache, (const CT*)v_cache,              \
                         block_tables, seq_lens, (short*)out, nullptr,        \
                         nullptr, kvh, bs, max_blocks, 0, scale);             \
  } while (0)
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
#undef LAUNCH_CT
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lds_paged_attention_split(
    const void* q, const void* k_cache, const void* v_cache,
    const int32_t* block_tables, const int32_t* seq_lens, void* out,
    float* part_o, float* part_ml, int n_seqs, int n_q_heads, int kvh, int bs,
    int head_dim, int max_blocks, int n_parts, int part_tokens, int kv_fp8,
    int chunk, float scale, hipStream_t stream) {
  if (n_seqs == 0) return hipSuccess;
  if (head_dim != D || n_parts > 64) return hipErrorInvalidValue;
  if (chunk != 256 && chunk != 512) return hipErrorInvalidValue;
  const int qpg = n_q_heads / kvh;
  dim3 grid(n_seqs, kvh, n_parts), block(NW * WAVE);
  dim3 cgrid(n_seqs, kvh), cblock(256);
#define LAUNCH_CT(QPG, CT)                                                    \
  do {                                                                        \
    if (chunk == 512)                                                         \
      hipLaunchKernelGGL((paged_attention_kernel<QPG, true, CT, 512>), grid,  \
                         block, 0, stream, (const short*)q,                   \
                         (const CT*)k_cache, (const CT*)v_cache,              \
                         block_tables, seq_lens, (short*)out, part_o,         \
                         part_ml, kvh, bs, max_blocks, part_tokens, scale);   \
    else                                                                      \
      hipLaunchKernelGGL((paged_attention_kernel<QPG, true, CT, 256>), grid,  \
                         block, 0, stream, (const short*)q,                   \
                         (const CT*)k_cache, (const CT*)v_cache,              \
                         block_tables, seq_lens, (short*)out, part_o,         \
                         part_ml, kvh, bs, max_blocks, part_tokens, scale);   \
  } while (0)
#define LAUNCH(QPG)                                                           \
  do {                                                                        \
    if (kv_fp8) LAUNCH_CT(QPG, unsigned char);                                \
    else LAUNCH_CT(QPG, short);                                               \
    hipLaunchKernelGGL((paged_attention_combine_kernel<QPG>), cgrid, cblock,  \
                       0, stream, part_o, part_ml, (short*)out, kvh,          \
                       n_parts);                                              \
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
