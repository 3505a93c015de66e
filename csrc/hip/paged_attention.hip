// paged_attention.hip — GQA decode attention over the paged KV pool (gfx950).
//
// The decode-step hot kernel of the per-GPU worker engine. Memory-bound: the
// whole point is streaming each sequence's K/V exactly once per step at
// near-HBM rate (guide Appendix B "Attention decode").
//
// Geometry: one workgroup (4 waves, 256 threads) per (sequence, kv_head).
// Q-heads in the GQA group (QPG = n_q_heads / n_kv_heads, <= 8) are scored
// together so K rows are read once for the whole group.
//   Phase A  each wave owns 64 tokens of a 256-token chunk; a lane streams
//            its token's K row (short8 loads) against the group's Q vectors
//            staged in LDS (broadcast reads) -> logits in LDS.
//   Phase B  online-softmax update per q-head (wave-parallel reductions).
//   Phase C  V accumulation: lane owns a dim pair, wave owns its 64 tokens;
//            V rows stream fully coalesced (64 lanes x 4 B = one 256 B row).
//   Final    cross-wave combine via LDS, normalize, bf16 store.
#include "hip_common.h"

namespace {

constexpr int D = 128;        // head_dim (Llama-3 family)
constexpr int CHUNK = 256;    // tokens per online-softmax chunk
constexpr int NW = 4;         // waves per workgroup
constexpr float NEG = -1e30f;

template <int QPG>
__global__ __launch_bounds__(NW * WAVE) void paged_attention_kernel(
    const short* __restrict__ q,        // [B, QH, D]
    const short* __restrict__ k_cache,  // [NB, KVH, BS, D]
    const short* __restrict__ v_cache,  // [NB, KVH, BS, D]
    const int32_t* __restrict__ block_tables,  // [B, max_blocks]
    const int32_t* __restrict__ seq_lens,      // [B]
    short* __restrict__ out,                   // [B, QH, D]
    int kvh, int bs, int max_blocks, float scale) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int qh0 = kh * QPG;
  const int n_q_heads = kvh * QPG;
  const int seq_len = seq_lens[b];

  __shared__ float q_lds[QPG][D];
  __shared__ float logits[QPG][CHUNK];
  __shared__ float m_sh[QPG], l_sh[QPG], alpha_sh[QPG];
  __shared__ float comb[NW][QPG][D];

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;

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

  for (int chunk0 = 0; chunk0 < seq_len; chunk0 += CHUNK) {
    const int n_t = min(CHUNK, seq_len - chunk0);
    // ---- Phase A: logits[h][t_local] ----
    {
      const int t_local = wave * WAVE + lane;
      float dot[QPG];
#pragma unroll
      for (int h = 0; h < QPG; ++h) dot[h] = 0.f;
      if (t_local < n_t) {
        const int t = chunk0 + t_local;
        const int64_t blk = bt[t / bs];
        const int row = t % bs;
        const short8* krow =
            (const short8*)(k_cache + (((blk * kvh + kh) * bs) + row) * D);
#pragma unroll 4
        for (int c = 0; c < D / 8; ++c) {
          short8 kv8 = krow[c];
          float kf[8];
#pragma unroll
          for (int j = 0; j < 8; ++j) kf[j] = bf16_to_f32(kv8[j]);
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
        logits[h][wave * WAVE + lane] = (t_local < n_t) ? dot[h] : NEG;
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
      const int t_base = wave * WAVE;
      const int t_cnt = min(WAVE, n_t - t_base);
      for (int i = 0; i < t_cnt; ++i) {
        const int t = chunk0 + t_base + i;
        const int64_t blk = bt[t / bs];
        const int row = t % bs;
        const int32_t* vrow =
            (const int32_t*)(v_cache + (((blk * kvh + kh) * bs) + row) * D);
        const int32_t pair = vrow[lane];  // 2 bf16, coalesced 256B row
        const float v0 = bf16_to_f32((short)(pair & 0xFFFF));
        const float v1 = bf16_to_f32((short)((pair >> 16) & 0xFFFF));
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

}  // namespace

extern "C" {

hipError_t lds_paged_attention(const void* q, const void* k_cache,
                               const void* v_cache, const int32_t* block_tables,
                               const int32_t* seq_lens, void* out, int n_seqs,
                               int n_q_heads, int kvh, int bs, int head_dim,
                               int max_blocks, float scale,
                               hipStream_t stream) {
  if (n_seqs == 0) return hipSuccess;
  if (head_dim != D) return hipErrorInvalidValue;
  const int qpg = n_q_heads / kvh;
  dim3 grid(n_seqs, kvh), block(NW * WAVE);
#define LAUNCH(QPG)                                                           \
  hipLaunchKernelGGL(paged_attention_kernel<QPG>, grid, block, 0, stream,     \
                     (const short*)q, (const short*)k_cache,                  \
                     (const short*)v_cache, block_tables, seq_lens,           \
                     (short*)out, kvh, bs, max_blocks, scale)
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
