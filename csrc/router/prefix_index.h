// prefix_index.h — node-local prefix-cache index: chained block hash ->
// {set of endpoints believed to hold that block} plus a per-endpoint LRU
// bounding memory, with eviction back out of the hash map.
//
// Capability parity with reference `approximateprefix/indexer.go:32-115`
// (hashToPods map + per-pod LRU with eviction callback) re-designed for one
// 8-GPU node: endpoints are small dense indices so the "pod set" is a single
// uint64 bitmask, the LRU is an intrusive doubly-linked list over a flat
// entry pool (no per-node allocation), and the whole index is guarded by a
// shared_mutex (reads = match scans dominate).
#pragma once
#include <cstdint>
#include <cstring>
#include <unordered_map>
#include <vector>
#include <deque>
#include <shared_mutex>
#include <mutex>

#include "../common/xxhash64.h"

namespace ldsr {

static constexpr int kMaxEndpoints = 64;

class PrefixIndex {
 public:
  explicit PrefixIndex(int64_t lru_capacity_per_endpoint)
      : cap_(lru_capacity_per_endpoint) {}

  struct Entry {
    uint64_t mask = 0;  // endpoints holding this block
  };

  // Per-endpoint LRU node (one per (endpoint, hash) residency).
  struct LruNode {
    uint64_t hash;
    int32_t prev = -1, next = -1;
    bool live = false;
  };

  struct LruList {
    std::vector<LruNode> pool;
    std::deque<int32_t> free;
    std::unordered_map<uint64_t, int32_t> pos;  // hash -> pool slot
    int32_t head = -1, tail = -1;               // head = most recent
    int64_t size = 0;

    void unlink(int32_t i) {
      LruNode& n = pool[i];
      if (n.prev >= 0) pool[n.prev].next = n.next; else head = n.next;
      if (n.next >= 0) pool[n.next].prev = n.prev; else tail = n.prev;
      n.prev = n.next = -1;
    }
    void push_front(int32_t i) {
      LruNode& n = pool[i];
      n.prev = -1; n.next = head;
      if (head >= 0) pool[head].prev = i;
      head = i;
      if (tail < 0) tail = i;
    }
  };

  // Record that `endpoint` now holds these blocks (most-recent first touch).
  // Returns number of evicted blocks.
  int64_t add(int endpoint, const uint64_t* hashes, int64_t n) {
    std::unique_lock lk(mu_);
    if (endpoint < 0 || endpoint >= kMaxEndpoints) return 0;
    LruList& lru = lrus_[endpoint];
    const uint64_t bit = 1ULL << endpoint;
    for (int64_t i = 0; i < n; ++i) {
      uint64_t h = hashes[i];
      auto it = lru.pos.find(h);
      if (it != lru.pos.end()) {
        lru.unlink(it->second);
        lru.push_front(it->second);
        continue;
      }
      int32_t slot;
      if (!lru.free.empty()) { slot = lru.free.front(); lru.free.pop_front(); }
      else { slot = (int32_t)lru.pool.size(); lru.pool.emplace_back(); }
      lru.pool[slot].hash = h;
      lru.pool[slot].live = true;
      lru.push_front(slot);
      lru.pos.emplace(h, slot);
      lru.size++;
      map_[h].mask |= bit;
    }
    // evict over capacity (oldest first)
    int64_t evicted = 0;
    while (lru.size > cap_ && lru.tail >= 0) {
      int32_t victim = lru.tail;
      uint64_t h = lru.pool[victim].hash;
      lru.unlink(victim);
      lru.pool[victim].live = false;
      lru.free.push_back(victim);
      lru.pos.erase(h);
      lru.size--;
      auto mit = map_.find(h);
      if (mit != map_.end()) {
        mit->second.mask &= ~bit;
        if (mit->second.mask == 0) map_.erase(mit);
      }
      evicted++;
    }
    return evicted;
  }

  // Longest-prefix match (reference `plugin.go:214-230` matchLongestPrefix):
  // walk the chain hashes in order; an endpoint's match count is the number
  // of consecutive leading blocks it holds. Stops early once no endpoint
  // holds the next block. Returns per-endpoint counts for `num_endpoints`.
  void match_longest(const uint64_t* hashes, int64_t n, int num_endpoints,
                     int32_t* out_counts /* size num_endpoints */) const {
    std::shared_lock lk(mu_);
    for (int e = 0; e < num_endpoints; ++e) out_counts[e] = 0;
    uint64_t active = (num_endpoints >= 64) ? ~0ULL
                                            : ((1ULL << num_endpoints) - 1);
    for (int64_t i = 0; i < n && active; ++i) {
      auto it = map_.find(hashes[i]);
      uint64_t mask = (it == map_.end()) ? 0 : it->second.mask;
      uint64_t survivors = active & mask;
      uint64_t dropped = active & ~mask;
      // endpoints still matching after block i get count i+1
      uint64_t m = survivors;
      while (m) {
        int e = __builtin_ctzll(m);
        out_counts[e] = (int32_t)(i + 1);
        m &= m - 1;
      }
      (void)dropped;
      active = survivors;
    }
  }

  void remove_endpoint(int endpoint) {
    std::unique_lock lk(mu_);
    if (endpoint < 0 || endpoint >= kMaxEndpoints) return;
    LruList& lru = lrus_[endpoint];
    const uint64_t bit = 1ULL << endpoint;
    for (auto& [h, slot] : lru.pos) {
      auto mit = map_.find(h);
      if (mit != map_.end()) {
        mit->second.mask &= ~bit;
        if (mit->second.mask == 0) map_.erase(mit);
      }
    }
    lru = LruList{};
  }

  int64_t size() const {
    std::shared_lock lk(mu_);
    return (int64_t)map_.size();
  }
  int64_t endpoint_size(int endpoint) const {
    std::shared_lock lk(mu_);
    return lrus_[endpoint].size;
  }
  void set_capacity(int64_t cap) {
    std::unique_lock lk(mu_);
    cap_ = cap;
  }
  int64_t capacity() const { return cap_; }

 private:
  mutable std::shared_mutex mu_;
  int64_t cap_;
  std::unordered_map<uint64_t, Entry> map_;
  LruList lrus_[kMaxEndpoints];
};

// Hash a token sequence into chained block hashes (shared definition with
// the gfx950 kernel — see csrc/common/xxhash64.h header comment).
inline int64_t hash_tokens(const int32_t* tokens, int64_t n_tokens,
                           int block_tokens, int64_t max_blocks,
                           uint64_t seed0, uint64_t* out_hashes) {
  // Only complete blocks are hashed (reference types.go:92-113 semantics).
  int64_t n_blocks = n_tokens / block_tokens;
  if (n_blocks > max_blocks) n_blocks = max_blocks;
  uint64_t prev = seed0;
  for (int64_t b = 0; b < n_blocks; ++b) {
    uint64_t c = block_content_hash(tokens + b * block_tokens, block_tokens, seed0);
    prev = chain_hash(c, prev);
    out_hashes[b] = prev;
  }
  return n_blocks;
}

inline uint64_t model_seed(const char* model, size_t model_len,
                           const char* salt, size_t salt_len) {
  // seed0 = XXH64(model || salt) — reference hashing.go:35-50 analog.
  std::vector<uint8_t> buf(model_len + salt_len);
  if (model_len) memcpy(buf.data(), model, model_len);
  if (salt_len) memcpy(buf.data() + model_len, salt, salt_len);
  return xxh64(buf.data(), buf.size(), 0);
}

}  // namespace ldsr
