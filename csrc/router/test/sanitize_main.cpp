// Sanitizer exerciser for the native router core (ASan+UBSan analog of the
// reference's `go test -race` tiers, Makefile:260,281). Drives
// PrefixIndex LRU/eviction, both flow queues, and ProfileRunner through
// randomized workloads under -fsanitize=address,undefined. Run via
// tools/sanitize_check.sh; tests/test_router_core.py::test_sanitizer_clean
// gates it in the CPU suite.
#include <cassert>
#include <cstdio>
#include <random>
#include <vector>

#include "../prefix_index.h"
#include "../queues.h"
#include "../scoring.h"

using namespace ldsr;

static void exercise_prefix_index(std::mt19937_64& rng) {
  PrefixIndex ix(64);  // small LRU: forces constant eviction churn
  std::uniform_int_distribution<uint64_t> h(0, 300);
  std::uniform_int_distribution<int> ep(0, 7), n(1, 32);
  for (int it = 0; it < 5000; ++it) {
    std::vector<uint64_t> hashes(n(rng));
    for (auto& x : hashes) x = h(rng);
    int e = ep(rng);
    switch (it % 5) {
      case 0: case 1: case 2:
        ix.add(e, hashes.data(), (int64_t)hashes.size());
        break;
      case 3: {
        std::vector<int32_t> counts(8);
        ix.match_longest(hashes.data(), (int64_t)hashes.size(), 8,
                         counts.data());
        for (int c : counts) assert(c >= 0 && c <= (int)hashes.size());
        break;
      }
      default:
        ix.remove_endpoint(e);
    }
    assert(ix.size() >= 0);
  }
  ix.set_capacity(4);   // shrink is applied lazily by the next add
  for (int e = 0; e < 8; ++e) {
    uint64_t one = 9999;
    ix.add(e, &one, 1);
    assert(ix.endpoint_size(e) <= 4);
  }
}

template <typename Q>
static void exercise_queue(std::mt19937_64& rng) {
  Q q;
  std::uniform_int_distribution<uint64_t> id(0, 200);
  std::uniform_real_distribution<double> key(0, 1);
  int64_t live_bytes = 0, live_len = 0;
  for (int it = 0; it < 20000; ++it) {
    uint64_t i = id(rng);
    int64_t nb = (int64_t)(key(rng) * 100);
    switch (it % 4) {
      case 0: case 1: {
        q.push(i, key(rng), nb);
        live_bytes += nb; live_len++;
        break;
      }
      case 2: {
        uint64_t out; int64_t ob;
        if (q.pop(&out, &ob)) { live_bytes -= ob; live_len--; }
        break;
      }
      default: {
        int64_t ob;
        if (q.remove(i, &ob)) { live_bytes -= ob; live_len--; }
      }
    }
    assert(q.len() == live_len && q.bytes() == live_bytes);
  }
}

static void exercise_runner(std::mt19937_64& rng) {
  ProfileRunner pr(123);
  std::uniform_real_distribution<float> f(0.f, 1.f);
  for (int it = 0; it < 2000; ++it) {
    int n = 1 + (int)(f(rng) * 63);
    Snapshot s;
    s.n = n;
    s.queue_depth.resize(n); s.running.resize(n); s.kv_usage.resize(n);
    s.inflight_tokens.resize(n); s.active_requests.resize(n);
    s.roles.resize(n);
    for (int i = 0; i < n; ++i) {
      s.queue_depth[i] = f(rng) * 100; s.running[i] = f(rng) * 100;
      s.kv_usage[i] = f(rng); s.inflight_tokens[i] = f(rng) * 1e5f;
      s.active_requests[i] = f(rng) * 50;
      s.roles[i] = (uint8_t)(1 + (int)(f(rng) * 6));
    }
    std::vector<ScorerSpec> specs = {
        {SC_QUEUE, 2.f, 0.f}, {SC_KV_UTIL, 1.f, 0.f},
        {SC_PREFIX, 3.f, 0.f}, {SC_LOAD_AWARE, 1.f, 64.f},
        {SC_TOKEN_LOAD, 1.f, 1e4f}};
    std::vector<int32_t> match(n);
    for (auto& x : match) x = (int32_t)(f(rng) * 8);
    auto r = pr.run(s, it % 2 ? ROLE_DECODE : 0, nullptr, specs,
                    match.data(), 8, nullptr, it % 3, 1 + it % 4);
    for (int idx : r.picks) assert(idx >= 0 && idx < n);
  }
}

int main() {
  std::mt19937_64 rng(42);
  exercise_prefix_index(rng);
  exercise_queue<ListQueue>(rng);
  exercise_queue<MaxMinHeap>(rng);
  exercise_runner(rng);
  std::puts("sanitize: OK");
  return 0;
}
