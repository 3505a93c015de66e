// prefix_kernels.hip — gfx950 kernels for the router's prefix-cache hot path.
//
// These are the MI355X-native replacement for the reference's CPU hot loops
// `approximateprefix/hashing.go:80-95` (chained xxhash over prompt blocks)
// and `approximateprefix/indexer.go:86-102` (hash -> pod-set longest-prefix
// match under RWMutex). Re-design for CDNA4:
//   * batched across the whole admission queue — one launch hashes and
//     matches EVERY queued request (the Go version is per-request serial);
//   * one 64-lane wave per request: block content hashes are lane-parallel,
//     only the 16-byte chain combine walks sequentially on lane 0;
//   * the hash->endpoint-mask table is a device-resident open-addressing
//     table (sized for 288 GB HBM budgets) probed directly by the match
//     kernel with the per-request hash chain staged in LDS.
// The chain definition is shared bit-exactly with the CPU core via
// csrc/common/xxhash64.h.
#include "hip_common.h"
#include "../common/xxhash64.h"

using ldsr::block_content_hash;
using ldsr::chain_hash;

namespace {

constexpr int MAX_BLOCKS_LDS = 512;  // per-request chain staging (4 KiB LDS)

// ---------------------------------------------------------------------------
// hash_prompts: one wave per request.
//  tokens:   flat int32 token buffer
//  offsets:  int64 [R+1] request start offsets into tokens
//  out:      uint64 [R, max_blocks] chained hashes
//  counts:   int32 [R] number of complete blocks hashed
// ---------------------------------------------------------------------------
__global__ void hash_prompts_kernel(const int32_t* __restrict__ tokens,
                                    const int64_t* __restrict__ offsets,
                                    int n_requests, int block_tokens,
                                    int max_blocks, uint64_t seed0,
                                    uint64_t* __restrict__ out,
                                    int32_t* __restrict__ counts) {
  __shared__ uint64_t content[MAX_BLOCKS_LDS];
  int req = blockIdx.x;
  if (req >= n_requests) return;
  int lane = threadIdx.x;  // blockDim.x == 64 (one wave)
  int64_t beg = offsets[req], end = offsets[req + 1];
  int n_blocks = (int)((end - beg) / block_tokens);
  if (n_blocks > max_blocks) n_blocks = max_blocks;
  if (lane == 0) counts[req] = n_blocks;
  // lane-parallel content hashes
  for (int b = lane; b < n_blocks; b += WAVE) {
    content[b] = block_content_hash(tokens + beg + (int64_t)b * block_tokens,
                                    block_tokens, seed0);
  }
  __syncthreads();
  // sequential chain on lane 0 (16-byte combines; cheap)
  if (lane == 0) {
    uint64_t prev = seed0;
    uint64_t* dst = out + (int64_t)req * max_blocks;
    for (int b = 0; b < n_blocks; ++b) {
      prev = chain_hash(content[b], prev);
      dst[b] = prev;
    }
  }
}

// ---------------------------------------------------------------------------
// Device hash table: open addressing, linear probing.
// keys[cap] (0 = empty; real hash 0 is remapped to 1), masks[cap].
// The host-side C++ PrefixIndex owns LRU policy; inserts/removals are
// mirrored here in batches.
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint64_t norm_key(uint64_t h) { return h ? h : 1; }

__global__ void table_insert_kernel(uint64_t* __restrict__ keys,
                                    unsigned long long* __restrict__ masks,
                                    uint32_t cap_mask,  // cap-1, cap = 2^k
                                    const uint64_t* __restrict__ hashes,
                                    int64_t n, unsigned long long bit) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t key = norm_key(hashes[i]);
  uint32_t slot = (uint32_t)key & cap_mask;
  for (uint32_t probe = 0; probe <= cap_mask; ++probe) {
    uint64_t prev = atomicCAS((unsigned long long*)&keys[slot], 0ULL,
                              (unsigned long long)key);
    if (prev == 0 || prev == key) {
      atomicOr(&masks[slot], bit);
      return;
    }
    slot = (slot + 1) & cap_mask;
  }
  // table full: drop (host monitors load factor and rebuilds)
}

__global__ void table_remove_kernel(const uint64_t* __restrict__ keys,
                                    unsigned long long* __restrict__ masks,
                                    uint32_t cap_mask,
                                    const uint64_t* __restrict__ hashes,
                                    int64_t n, unsigned long long bit) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t key = norm_key(hashes[i]);
  uint32_t slot = (uint32_t)key & cap_mask;
  for (uint32_t probe = 0; probe <= cap_mask; ++probe) {
    uint64_t k = keys[slot];
    if (k == 0) return;  // not present
    if (k == key) {
      atomicAnd(&masks[slot], ~bit);
      return;
    }
    slot = (slot + 1) & cap_mask;
  }
}

__device__ __forceinline__ unsigned long long table_lookup(
    const uint64_t* __restrict__ keys,
    const unsigned long long* __restrict__ masks, uint32_t cap_mask,
    uint64_t key) {
  uint32_t slot = (uint32_t)key & cap_mask;
  for (uint32_t probe = 0; probe <= cap_mask; ++probe) {
    uint64_t k = keys[slot];
    if (k == 0) return 0;
    if (k == key) return masks[slot];
    slot = (slot + 1) & cap_mask;
  }
  return 0;
}

// ---------------------------------------------------------------------------
// match_longest: one wave per request. Probes the table for each chain hash
// (lane-parallel, staged through LDS) then computes the per-endpoint longest
// consecutive prefix exactly like the CPU path.
//  hashes:  uint64 [R, max_blocks]
//  counts:  int32 [R]
//  out:     int32 [R, n_endpoints] consecutive matched blocks per endpoint
// ---------------------------------------------------------------------------
__global__ void match_longest_kernel(const uint64_t* __restrict__ keys,
                                     const unsigned long long* __restrict__ masks,
                                     uint32_t cap_mask,
                                     const uint64_t* __restrict__ hashes,
                                     const int32_t* __restrict__ counts,
                                     int n_requests, int max_blocks,
                                     int n_endpoints,
                                     int32_t* __restrict__ out) {
  __shared__ unsigned long long block_mask[MAX_BLOCKS_LDS];
  int req = blockIdx.x;
  if (req >= n_requests) return;
  int lane = threadIdx.x;
  int n_blocks = counts[req];
  if (n_blocks > max_blocks) n_blocks = max_blocks;
  const uint64_t* h = hashes + (int64_t)req * max_blocks;
  for (int b = lane; b < n_blocks; b += WAVE) {
    block_mask[b] = table_lookup(keys, masks, cap_mask, norm_key(h[b]));
  }
  __syncthreads();
  if (lane == 0) {
    unsigned long long active =
        (n_endpoints >= 64) ? ~0ULL : ((1ULL << n_endpoints) - 1);
    int32_t* dst = out + (int64_t)req * n_endpoints;
    for (int e = 0; e < n_endpoints; ++e) dst[e] = 0;
    for (int b = 0; b < n_blocks && active; ++b) {
      unsigned long long surv = active & block_mask[b];
      unsigned long long m = surv;
      while (m) {
        int e = __builtin_ctzll(m);
        dst[e] = b + 1;
        m &= m - 1;
      }
      active = surv;
    }
  }
}

}  // namespace

// ---- host-side launchers (called from the torch binding) ----

extern "C" {

hipError_t lds_hash_prompts(const int32_t* tokens, const int64_t* offsets,
                            int n_requests, int block_tokens, int max_blocks,
                            uint64_t seed0, uint64_t* out, int32_t* counts,
                            hipStream_t stream) {
  if (n_requests == 0) return hipSuccess;
  hipLaunchKernelGGL(hash_prompts_kernel, dim3(n_requests), dim3(WAVE), 0,
                     stream, tokens, offsets, n_requests, block_tokens,
                     max_blocks, seed0, out, counts);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lds_table_update(uint64_t* keys, unsigned long long* masks,
                            uint32_t cap_mask, const uint64_t* hashes,
                            int64_t n, int endpoint, int is_remove,
                            hipStream_t stream) {
  if (n == 0) return hipSuccess;
  unsigned long long bit = 1ULL << endpoint;
  int threads = 256;
  int blocks = (int)((n + threads - 1) / threads);
  if (is_remove) {
    hipLaunchKernelGGL(table_remove_kernel, dim3(blocks), dim3(threads), 0,
                       stream, keys, masks, cap_mask, hashes, n, bit);
  } else {
    hipLaunchKernelGGL(table_insert_kernel, dim3(blocks), dim3(threads), 0,
                       stream, keys, masks, cap_mask, hashes, n, bit);
  }
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lds_match_longest(const uint64_t* keys,
                             const unsigned long long* masks,
                             uint32_t cap_mask, const uint64_t* hashes,
                             const int32_t* counts, int n_requests,
                             int max_blocks, int n_endpoints, int32_t* out,
                             hipStream_t stream) {
  if (n_requests == 0) return hipSuccess;
  hipLaunchKernelGGL(match_longest_kernel, dim3(n_requests), dim3(WAVE), 0,
                     stream, keys, masks, cap_mask, hashes, counts, n_requests,
                     max_blocks, n_endpoints, out);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
