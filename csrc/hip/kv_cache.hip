// kv_cache.hip — paged-KV pool maintenance kernels for gfx950.
//
// Pool layout (per GPU, sized for the 288 GB HBM3E budget):
//   kv_pool: [n_layers, 2(K/V), n_blocks, n_kv_heads, block_size, head_dim] bf16
// One (layer, K/V, block) = a contiguous [n_kv_heads, block_size, head_dim]
// chunk whose per-head [block_size x head_dim] rows are contiguous — the
// MFMA-tile-aligned layout consumed zero-repack by both the decode attention
// kernel and the xGMI transfer engine.
#include "hip_common.h"

namespace {

// Scatter the freshly projected K/V of each token into its pool slot.
//  k_new/v_new: [T, KVH, D] bf16; slot_mapping: [T] int64 (block*BS + row)
//  k_cache/v_cache: [NB, KVH, BS, D] bf16 (one layer's slice)
template <typename CT>
__global__ void reshape_and_cache_kernel(const short* __restrict__ k_new,
                                         const short* __restrict__ v_new,
                                         CT* __restrict__ k_cache,
                                         CT* __restrict__ v_cache,
                                         const int64_t* __restrict__ slots,
                                         int n_tokens, int kvh, int bs, int d) {
  const int t = blockIdx.x;
  if (t >= n_tokens) return;
  int64_t slot = slots[t];
  if (slot < 0) return;  // padding token
  int64_t block = slot / bs, row = slot % bs;
  // per (head, 16B chunk): blockDim.x = kvh * d/8 capped; grid-stride inside
  const int chunks_per_head = d / 8;
  const int total = kvh * chunks_per_head;
  for (int i = threadIdx.x; i < total; i += blockDim.x) {
    int h = i / chunks_per_head, c = i % chunks_per_head;
    const short8* src_k = (const short8*)(k_new + ((int64_t)t * kvh + h) * d);
    const short8* src_v = (const short8*)(v_new + ((int64_t)t * kvh + h) * d);
    int64_t dst_off = (((block * kvh + h) * bs) + row) * d;
    store_kv8(k_cache + dst_off + c * 8, src_k[c]);
    store_kv8(v_cache + dst_off + c * 8, src_v[c]);
  }
}

// Gather whole KV blocks (all layers, K+V) into a contiguous staging buffer
// for an xGMI send:  staging: [n_sel, L, 2, KVH, BS, D].
// pool: [L, 2, NB, KVH, BS, D]. block_ids: [n_sel].
__global__ void gather_blocks_kernel(const short8* __restrict__ pool,
                                     short8* __restrict__ staging,
                                     const int32_t* __restrict__ block_ids,
                                     int n_sel, int n_layers, int64_t nb,
                                     int64_t block_elems8 /* KVH*BS*D/8 */) {
  // one unit = one short8 chunk of one (sel, layer, kv) block copy
  const int64_t per_sel = (int64_t)n_layers * 2 * block_elems8;
  const int64_t total = per_sel * n_sel;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t sel = i / per_sel;
    int64_t rem = i % per_sel;
    int64_t lkv = rem / block_elems8;       // layer*2 + kv
    int64_t off = rem % block_elems8;
    int64_t src_block = block_ids[sel];
    staging[i] = pool[(lkv * nb + src_block) * block_elems8 + off];
  }
}

// Scatter a staging buffer (same layout) into this GPU's pool blocks.
__global__ void scatter_blocks_kernel(short8* __restrict__ pool,
                                      const short8* __restrict__ staging,
                                      const int32_t* __restrict__ block_ids,
                                      int n_sel, int n_layers, int64_t nb,
                                      int64_t block_elems8) {
  const int64_t per_sel = (int64_t)n_layers * 2 * block_elems8;
  const int64_t total = per_sel * n_sel;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t sel = i / per_sel;
    int64_t rem = i % per_sel;
    int64_t lkv = rem / block_elems8;
    int64_t off = rem % block_elems8;
    int64_t dst_block = block_ids[sel];
    pool[(lkv * nb + dst_block) * block_elems8 + off] = staging[i];
  }
}

// Direct peer-to-peer block copy: gather blocks `src_ids` straight out of a
// PEER GPU's pool (mapped via hipIpcOpenMemHandle — loads travel over the
// direct xGMI link) into this GPU's pool blocks `dst_ids`. One-sided pull:
// no staging buffer, no send/recv rendezvous (SURVEY.md §5.8 names
// hipMemcpyPeerAsync; a gather kernel over the mapped peer pointer is the
// same xGMI path but handles the non-contiguous block list in one launch).
__global__ void copy_blocks_peer_kernel(const short8* __restrict__ src_pool,
                                        short8* __restrict__ dst_pool,
                                        const int32_t* __restrict__ src_ids,
                                        const int32_t* __restrict__ dst_ids,
                                        int n_sel, int n_layers,
                                        int64_t src_nb, int64_t dst_nb,
                                        int64_t block_elems8) {
  const int64_t per_sel = (int64_t)n_layers * 2 * block_elems8;
  const int64_t total = per_sel * n_sel;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t sel = i / per_sel;
    int64_t rem = i % per_sel;
    int64_t lkv = rem / block_elems8;
    int64_t off = rem % block_elems8;
    dst_pool[(lkv * dst_nb + dst_ids[sel]) * block_elems8 + off] =
        src_pool[(lkv * src_nb + src_ids[sel]) * block_elems8 + off];
  }
}

}  // namespace

extern "C" {

hipError_t lds_reshape_and_cache(const void* k_new, const void* v_new,
                                 void* k_cache, void* v_cache,
                                 const int64_t* slots, int n_tokens, int kvh,
                                 int bs, int d, int kv_fp8,
                                 hipStream_t stream) {
  if (n_tokens == 0) return hipSuccess;
  int threads = kvh * (d / 8);
  if (threads > 256) threads = 256;
  if (kv_fp8) {
    hipLaunchKernelGGL(reshape_and_cache_kernel<unsigned char>,
                       dim3(n_tokens), dim3(threads), 0, stream,
                       (const short*)k_new, (const short*)v_new,
                       (unsigned char*)k_cache, (unsigned char*)v_cache,
                       slots, n_tokens, kvh, bs, d);
  } else {
    hipLaunchKernelGGL(reshape_and_cache_kernel<short>, dim3(n_tokens),
                       dim3(threads), 0, stream, (const short*)k_new,
                       (const short*)v_new, (short*)k_cache, (short*)v_cache,
                       slots, n_tokens, kvh, bs, d);
  }
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lds_gather_blocks(const void* pool, void* staging,
                             const int32_t* block_ids, int n_sel, int n_layers,
                             int64_t n_blocks, int64_t block_bytes,
                             int is_scatter, hipStream_t stream) {
  // element-size agnostic: the kernels move 16-byte short8 units;
  // block_bytes is one (layer, K/V) block's bytes (must be 16-aligned)
  int64_t block_elems = block_bytes / 2;
  if (n_sel == 0) return hipSuccess;
  int64_t total = (int64_t)n_sel * n_layers * 2 * (block_elems / 8);
  int threads = 256;
  int blocks = (int)((total + threads - 1) / threads);
  if (blocks > 4096) blocks = 4096;
  if (is_scatter) {
    hipLaunchKernelGGL(scatter_blocks_kernel, dim3(blocks), dim3(threads), 0,
                       stream, (short8*)pool, (const short8*)staging, block_ids,
                       n_sel, n_layers, n_blocks, block_elems / 8);
  } else {
    hipLaunchKernelGGL(gather_blocks_kernel, dim3(blocks), dim3(threads), 0,
                       stream, (const short8*)pool, (short8*)staging, block_ids,
                       n_sel, n_layers, n_blocks, block_elems / 8);
  }
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lds_copy_blocks_peer(const void* src_pool, void* dst_pool,
                                const int32_t* src_ids, const int32_t* dst_ids,
                                int n_sel, int n_layers, int64_t src_nb,
                                int64_t dst_nb, int64_t block_bytes,
                                hipStream_t stream) {
  if (n_sel == 0) return hipSuccess;
  int64_t block_elems8 = block_bytes / 16;
  int64_t total = (int64_t)n_sel * n_layers * 2 * block_elems8;
  int threads = 256;
  int blocks = (int)((total + threads - 1) / threads);
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(copy_blocks_peer_kernel, dim3(blocks), dim3(threads), 0,
                     stream, (const short8*)src_pool, (short8*)dst_pool,
                     src_ids, dst_ids, n_sel, n_layers, src_nb, dst_nb,
                     block_elems8);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
