// kv_bw_probe.hip — what bandwidth does the decode-attention ACCESS PATTERN
// alone sustain? Same geometry as paged_attention_kernel (WG per
// (seq, kv_head), per-lane K-row streaming + row-iterated V reads), but no
// softmax/FMA work — just a checksum. Run on a GPU box:
//   hipcc --offload-arch=gfx950 tools/kv_bw_probe.hip -o /tmp/kvbw && /tmp/kvbw
#include "../csrc/hip/hip_common.h"
#include <cstdio>
#include <vector>

constexpr int D = 128, BS = 16, KVH = 8, NW = 4;

// variant 0: phase-A style (lane owns a row, 16x16B loads, 8 deep)
// variant 1: phase-C style (64 lanes x 4B cover a row, 8 rows deep)
template <int V>
__global__ __launch_bounds__(256) void probe(
    const short* __restrict__ kc, const int32_t* __restrict__ bt,
    int max_blocks, int seq_len, float* out) {
  const int b = blockIdx.x, kh = blockIdx.y;
  const int tid = threadIdx.x, wave = tid / 64, lane = tid % 64;
  const int32_t* my_bt = bt + (int64_t)b * max_blocks;
  float acc = 0.f;
  for (int chunk0 = 0; chunk0 < seq_len; chunk0 += 256) {
    const int n_t = min(256, seq_len - chunk0);
    if (V == 0) {
      const int t_local = wave * 64 + lane;
      if (t_local < n_t) {
        const int t = chunk0 + t_local;
        const short8* row = (const short8*)(
            kc + (((int64_t)my_bt[t / BS] * KVH + kh) * BS + t % BS) * D);
#pragma unroll 8
        for (int c = 0; c < 16; ++c) {
          short8 v = row[c];
          acc += (float)v[0] + (float)v[7];
        }
      }
    } else {
      const int t_base = wave * 64;
      const int t_cnt = min(64, n_t - t_base);
      for (int i = 0; i + 8 <= t_cnt; i += 8) {
        int32_t pairs[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int t = chunk0 + t_base + i + j;
          pairs[j] = ((const int32_t*)(
              kc + (((int64_t)my_bt[t / BS] * KVH + kh) * BS + t % BS) * D))
              [lane];
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) acc += (float)(short)(pairs[j] & 0xFFFF);
      }
    }
  }
  if (acc == 12345.678f) out[0] = acc;  // never true; defeats DCE
}

int main() {
  const int B = 128, seq = 1152;
  const int max_blocks = (seq + BS - 1) / BS;
  const int64_t nb = (int64_t)B * max_blocks + 1;
  short* kc;
  int32_t* bt;
  float* out;
  hipMalloc(&kc, nb * KVH * BS * D * 2);
  hipMalloc(&bt, B * max_blocks * 4);
  hipMalloc(&out, 4);
  std::vector<int32_t> hbt(B * max_blocks);
  for (size_t i = 0; i < hbt.size(); ++i) hbt[i] = (int32_t)(i + 1);
  hipMemcpy(bt, hbt.data(), hbt.size() * 4, hipMemcpyHostToDevice);
  hipMemset(kc, 0x3f, nb * KVH * BS * D * 2);
  double bytes = (double)B * seq * KVH * D * 2;  // one stream (K only)
  for (int v = 0; v < 2; ++v) {
    for (int rep = 0; rep < 3; ++rep) {
      hipEvent_t e0, e1;
      hipEventCreate(&e0);
      hipEventCreate(&e1);
      hipEventRecord(e0);
      for (int it = 0; it < 30; ++it) {
        if (v == 0)
          hipLaunchKernelGGL(probe<0>, dim3(B, KVH), dim3(256), 0, 0, kc, bt,
                             max_blocks, seq, out);
        else
          hipLaunchKernelGGL(probe<1>, dim3(B, KVH), dim3(256), 0, 0, kc, bt,
                             max_blocks, seq, out);
      }
      hipEventRecord(e1);
      hipEventSynchronize(e1);
      float ms;
      hipEventElapsedTime(&ms, e0, e1);
      double tbps = bytes * 30 / (ms / 1e3) / 1e12;
      if (rep == 2)
        printf("variant %d (%s): %7.1f us/iter  %5.2f TB/s\n", v,
               v == 0 ? "phaseA rows" : "phaseC cols", ms * 1000 / 30, tbps);
    }
  }
  return 0;
}
