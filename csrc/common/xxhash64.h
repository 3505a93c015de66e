// xxhash64.h — self-contained XXH64 implementation (public-domain algorithm,
// implemented from the xxHash specification) usable from host C++ and HIP
// device code. This is the single definition of the prefix-cache block hash
// chain used by the router: the CPU path (csrc/router) and the gfx950 kernels
// (csrc/hip/prefix_hash.hip) both include this header, so CPU==GPU bitwise.
//
// Capability parity: reference `approximateprefix/hashing.go:35-99` chains
// xxhash64 over prompt character blocks. We chain over token-id blocks
// (tokens are already available in-process) with a parallel-friendly split:
//   content_i = XXH64(le32(token_ids of block i), seed = seed0)
//   h_i       = XXH64(le64(content_i) || le64(h_{i-1}), seed = 0),  h_{-1} = seed0
//   seed0     = XXH64(model_name_bytes || salt_bytes, seed = 0)
// The content hashes are independent (computed wave-parallel on gfx950); only
// the 16-byte chain combine is sequential, which keeps the GPU scan fast.
#pragma once
#include <cstdint>
#include <cstddef>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define XXH_HD __host__ __device__ __forceinline__
#else
#define XXH_HD inline
#endif

namespace ldsr {

static constexpr uint64_t PRIME64_1 = 0x9E3779B185EBCA87ULL;
static constexpr uint64_t PRIME64_2 = 0xC2B2AE3D27D4EB4FULL;
static constexpr uint64_t PRIME64_3 = 0x165667B19E3779F9ULL;
static constexpr uint64_t PRIME64_4 = 0x85EBCA77C2B2AE63ULL;
static constexpr uint64_t PRIME64_5 = 0x27D4EB2F165667C5ULL;

XXH_HD uint64_t xxh_rotl64(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }

XXH_HD uint64_t xxh_round(uint64_t acc, uint64_t input) {
  acc += input * PRIME64_2;
  acc = xxh_rotl64(acc, 31);
  acc *= PRIME64_1;
  return acc;
}

XXH_HD uint64_t xxh_merge_round(uint64_t acc, uint64_t val) {
  val = xxh_round(0, val);
  acc ^= val;
  acc = acc * PRIME64_1 + PRIME64_4;
  return acc;
}

XXH_HD uint64_t xxh_read64(const uint8_t* p) {
  // unaligned little-endian read, byte-assembled (safe on host and device)
  uint64_t v = 0;
  for (int i = 7; i >= 0; --i) v = (v << 8) | p[i];
  return v;
}

XXH_HD uint32_t xxh_read32(const uint8_t* p) {
  return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) | ((uint32_t)p[3] << 24);
}

XXH_HD uint64_t xxh64(const void* data, size_t len, uint64_t seed) {
  const uint8_t* p = (const uint8_t*)data;
  const uint8_t* end = p + len;
  uint64_t h;
  if (len >= 32) {
    uint64_t v1 = seed + PRIME64_1 + PRIME64_2;
    uint64_t v2 = seed + PRIME64_2;
    uint64_t v3 = seed + 0;
    uint64_t v4 = seed - PRIME64_1;
    const uint8_t* limit = end - 32;
    do {
      v1 = xxh_round(v1, xxh_read64(p)); p += 8;
      v2 = xxh_round(v2, xxh_read64(p)); p += 8;
      v3 = xxh_round(v3, xxh_read64(p)); p += 8;
      v4 = xxh_round(v4, xxh_read64(p)); p += 8;
    } while (p <= limit);
    h = xxh_rotl64(v1, 1) + xxh_rotl64(v2, 7) + xxh_rotl64(v3, 12) + xxh_rotl64(v4, 18);
    h = xxh_merge_round(h, v1);
    h = xxh_merge_round(h, v2);
    h = xxh_merge_round(h, v3);
    h = xxh_merge_round(h, v4);
  } else {
    h = seed + PRIME64_5;
  }
  h += (uint64_t)len;
  while (p + 8 <= end) {
    h ^= xxh_round(0, xxh_read64(p));
    h = xxh_rotl64(h, 27) * PRIME64_1 + PRIME64_4;
    p += 8;
  }
  if (p + 4 <= end) {
    h ^= (uint64_t)xxh_read32(p) * PRIME64_1;
    h = xxh_rotl64(h, 23) * PRIME64_2 + PRIME64_3;
    p += 4;
  }
  while (p < end) {
    h ^= (*p) * PRIME64_5;
    h = xxh_rotl64(h, 11) * PRIME64_1;
    ++p;
  }
  h ^= h >> 33;
  h *= PRIME64_2;
  h ^= h >> 29;
  h *= PRIME64_3;
  h ^= h >> 32;
  return h;
}

// ---- prefix-cache chain definition (shared CPU/GPU) ----

// content hash of one token block (token ids as little-endian u32 stream)
XXH_HD uint64_t block_content_hash(const int32_t* tokens, int n, uint64_t seed0) {
  // int32 tokens are already little-endian in memory on both host and gfx950
  return xxh64(tokens, (size_t)n * 4, seed0);
}

// chain combine: h_i = XXH64(le64(content_i) || le64(h_prev), seed=0)
XXH_HD uint64_t chain_hash(uint64_t content, uint64_t h_prev) {
  uint64_t buf[2] = {content, h_prev};
  return xxh64(buf, 16, 0);
}

}  // namespace ldsr
