// scoring.h — native scheduler hot loop: filter -> weighted score -> pick
// over the node's endpoint snapshots.
//
// Capability parity: reference `pkg/epp/scheduling/scheduler_profile.go:117-192`
// (filters sequentially, weighted scorers accumulate with scores clamped to
// [0,1], single picker) and the built-in scorer formulas of
// `pkg/epp/framework/plugins/scheduling/scorer/*` — re-designed as a batch
// evaluation over dense per-endpoint arrays (the node has <=64 GPU-role
// endpoints, so a scheduling cycle is a handful of vectorizable passes, not
// a plugin-object walk). Python-side plugins can still contribute via
// `extra` additive pre-weighted score arrays.
#pragma once
#include <cstdint>
#include <vector>
#include <random>
#include <cmath>
#include <algorithm>

namespace ldsr {

// Role bits (mirror of llm-d.ai/role label values; roles.go:9-48)
enum RoleBit : uint8_t {
  ROLE_DECODE = 1,
  ROLE_PREFILL = 2,
  ROLE_ENCODE = 4,
};

enum ScorerKind : int32_t {
  SC_QUEUE = 0,           // min-max normalized waiting-queue depth (inverted)
  SC_KV_UTIL = 1,         // 1 - kvCacheUsage
  SC_PREFIX = 2,          // match_blocks / total_blocks
  SC_RUNNING = 3,         // min-max normalized running count (inverted)
  SC_LOAD_AWARE = 4,      // 0.5 at empty queue -> 0 at threshold (param)
  SC_TOKEN_LOAD = 5,      // 1 - inflight_tokens / threshold (param)
  SC_ACTIVE_REQUEST = 6,  // min-max normalized EPP-tracked in-flight (inverted)
};

enum PickerKind : int32_t {
  PK_MAX_SCORE = 0,
  PK_RANDOM = 1,
  PK_WEIGHTED_RANDOM = 2,  // A-Res reservoir: key = U^(1/score)
};

struct ScorerSpec {
  int32_t kind;
  float weight;
  float param;   // threshold for LOAD_AWARE / TOKEN_LOAD; idleThreshold for
                 // ACTIVE_REQUEST; unused otherwise
  float param2;  // maxBusyScore for ACTIVE_REQUEST ((0,1]; <=0 -> 1.0)
};

struct Snapshot {
  // Dense per-endpoint state, refreshed by the datalayer each cycle.
  int n = 0;
  std::vector<float> queue_depth;
  std::vector<float> running;
  std::vector<float> kv_usage;        // [0,1]
  std::vector<float> inflight_tokens;
  std::vector<float> active_requests; // router-tracked in-flight
  std::vector<uint8_t> roles;         // RoleBit mask per endpoint
};

struct ProfileResult {
  std::vector<int32_t> picks;        // chosen endpoint indices, best first
  std::vector<float> scores;         // final weighted score per endpoint (-1 = filtered)
};

inline void minmax_inverted(const float* v, const uint8_t* alive, int n, float* out) {
  float lo = 1e30f, hi = -1e30f;
  for (int i = 0; i < n; ++i) if (alive[i]) { lo = std::min(lo, v[i]); hi = std::max(hi, v[i]); }
  float span = hi - lo;
  for (int i = 0; i < n; ++i) {
    if (!alive[i]) { out[i] = 0.f; continue; }
    out[i] = (span <= 0.f) ? 1.f : 1.f - (v[i] - lo) / span;
  }
}

class ProfileRunner {
 public:
  explicit ProfileRunner(uint64_t seed) : rng_(seed) {}

  // match_blocks: per-endpoint matched prefix blocks for THIS request
  // extra: optional pre-weighted additive scores from python-side plugins
  ProfileResult run(const Snapshot& s,
                    uint8_t role_filter,            // 0 = keep all
                    const uint8_t* candidate_mask,  // nullptr = all (subset hints)
                    const std::vector<ScorerSpec>& scorers,
                    const int32_t* match_blocks, int32_t total_blocks,
                    const float* extra,             // nullptr or size n
                    int32_t picker, int32_t max_endpoints) {
    const int n = s.n;
    ProfileResult r;
    r.scores.assign(n, -1.f);
    std::vector<uint8_t> alive(n, 1);
    for (int i = 0; i < n; ++i) {
      if (role_filter && !(s.roles[i] & role_filter)) alive[i] = 0;
      if (candidate_mask && !candidate_mask[i]) alive[i] = 0;
    }
    int n_alive = 0;
    for (int i = 0; i < n; ++i) n_alive += alive[i];
    if (n_alive == 0) return r;

    std::vector<float> acc(n, 0.f), tmp(n, 0.f);
    float total_w = 0.f;
    for (const auto& sp : scorers) {
      total_w += sp.weight;
      switch (sp.kind) {
        case SC_QUEUE: minmax_inverted(s.queue_depth.data(), alive.data(), n, tmp.data()); break;
        case SC_RUNNING: minmax_inverted(s.running.data(), alive.data(), n, tmp.data()); break;
        case SC_ACTIVE_REQUEST: {
          // activerequest/active_request.go:160-168: idle (count <=
          // idleThreshold) pins 1.0; busy scales (max-c)/max * maxBusyScore
          float idle_thr = std::max(0.f, sp.param);
          float busy_max = (sp.param2 > 0.f && sp.param2 <= 1.f) ? sp.param2 : 1.f;
          float hi = 0.f;
          for (int i = 0; i < n; ++i)
            if (alive[i]) hi = std::max(hi, s.active_requests[i]);
          for (int i = 0; i < n; ++i) {
            float c = s.active_requests[i];
            tmp[i] = (c <= idle_thr) ? 1.f
                     : (hi > 0.f ? (hi - c) / hi * busy_max : 1.f);
          }
          break;
        }
        case SC_KV_UTIL:
          for (int i = 0; i < n; ++i) tmp[i] = 1.f - s.kv_usage[i];
          break;
        case SC_PREFIX:
          for (int i = 0; i < n; ++i)
            tmp[i] = (total_blocks > 0 && match_blocks) ? (float)match_blocks[i] / (float)total_blocks : 0.f;
          break;
        case SC_LOAD_AWARE: {
          float thr = sp.param > 0 ? sp.param : 128.f;
          for (int i = 0; i < n; ++i)
            tmp[i] = std::max(0.f, 0.5f * (1.f - s.queue_depth[i] / thr));
          break;
        }
        case SC_TOKEN_LOAD: {
          float thr = sp.param > 0 ? sp.param : 1e6f;
          for (int i = 0; i < n; ++i)
            tmp[i] = std::max(0.f, 1.f - s.inflight_tokens[i] / thr);
          break;
        }
        default:
          for (int i = 0; i < n; ++i) tmp[i] = 0.f;
      }
      for (int i = 0; i < n; ++i) {
        // clamp to [0,1] per scheduler_profile.go:194-202
        float v = std::min(1.f, std::max(0.f, tmp[i]));
        if (alive[i]) acc[i] += sp.weight * v;
      }
    }
    if (extra) {
      for (int i = 0; i < n; ++i) if (alive[i]) acc[i] += extra[i];
    }
    for (int i = 0; i < n; ++i) if (alive[i]) r.scores[i] = acc[i];

    // pick
    int k = std::max(1, (int)max_endpoints);
    std::vector<int> cand;
    cand.reserve(n_alive);
    for (int i = 0; i < n; ++i) if (alive[i]) cand.push_back(i);
    std::uniform_real_distribution<double> uni(0.0, 1.0);
    if (picker == PK_RANDOM) {
      std::shuffle(cand.begin(), cand.end(), rng_);
    } else if (picker == PK_WEIGHTED_RANDOM) {
      bool any_pos = false;
      for (int i : cand) if (r.scores[i] > 0) { any_pos = true; break; }
      std::vector<std::pair<double, int>> keyed;
      keyed.reserve(cand.size());
      for (int i : cand) {
        double u = uni(rng_);
        double key = any_pos
            ? ((r.scores[i] > 0) ? std::pow(u, 1.0 / r.scores[i]) : -1.0)
            : u;  // uniform fallback when all scores <= 0 (weightedrandom/README)
        keyed.emplace_back(key, i);
      }
      std::sort(keyed.begin(), keyed.end(), [](auto& a, auto& b) { return a.first > b.first; });
      cand.clear();
      for (auto& [key, i] : keyed) cand.push_back(i);
    } else {  // PK_MAX_SCORE: sort by score desc, random tie-break
      std::vector<std::pair<double, int>> keyed;
      keyed.reserve(cand.size());
      for (int i : cand) keyed.emplace_back((double)r.scores[i] + 1e-9 * uni(rng_), i);
      std::sort(keyed.begin(), keyed.end(), [](auto& a, auto& b) { return a.first > b.first; });
      cand.clear();
      for (auto& [key, i] : keyed) cand.push_back(i);
    }
    if ((int)cand.size() > k) cand.resize(k);
    r.picks.assign(cand.begin(), cand.end());
    return r;
  }

 private:
  std::mt19937_64 rng_;
};

}  // namespace ldsr
