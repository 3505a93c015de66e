// mfma_probe.hip — empirical verification of gfx950 MFMA fragment layouts.
//
// Build/run ON a GPU box:
//   hipcc --offload-arch=gfx950 tools/mfma_probe.hip -o /tmp/mfma_probe
//   /tmp/mfma_probe
//
// Probe 1 (f32 32x32x2): A/B lane maps are documented (guide §3: lane l
// holds A[l&31][l>>5] / B[l>>5][l&31]); uses them to verify the C/D map.
// Probe 2 (bf16 32x32x16): tries candidate A/B fill maps and reports which
// combination reproduces A@B under the verified C map.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

__global__ void probe_f32(const float* A, const float* B, float* raw) {
  int l = threadIdx.x;
  float a = A[(l & 31) * 2 + (l >> 5)];     // A[i][k], i=l&31, k=l>>5
  float b = B[(l >> 5) * 32 + (l & 31)];    // B[k][j]
  f32x16 c = (f32x16)(0.f);
  c = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, c, 0, 0, 0);
  for (int r = 0; r < 16; ++r) raw[l * 16 + r] = c[r];
}

// bf16 probe: fill fragments under candidate map AV/BV
//   AV0: a_frag[e] = A[l&31][8*(l>>5)+e]   BV0: b_frag[e] = B[8*(l>>5)+e][l&31]
//   AV1: a_frag[e] = A[l&31][2*e+(l>>5)]   BV1: b_frag[e] = B[2*e+(l>>5)][l&31]
template <int AV, int BV>
__global__ void probe_bf16(const float* A, const float* B, float* raw) {
  int l = threadIdx.x;
  bf16x8 a, b;
  for (int e = 0; e < 8; ++e) {
    int ka = (AV == 0) ? 8 * (l >> 5) + e : 2 * e + (l >> 5);
    int kb = (BV == 0) ? 8 * (l >> 5) + e : 2 * e + (l >> 5);
    a[e] = (__bf16)A[(l & 31) * 16 + ka];
    b[e] = (__bf16)B[kb * 32 + (l & 31)];
  }
  f32x16 c = (f32x16)(0.f);
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 16; ++r) raw[l * 16 + r] = c[r];
}

static int row_c0(int r, int hi) { return (r & 3) + 8 * (r >> 2) + 4 * hi; }
static int row_c1(int r, int hi) { return r + 16 * hi; }
static int row_c2(int r, int hi) { return 2 * r + hi; }

static bool check(const float* raw, const float* ref, int cv, const char* tag) {
  int bad = 0;
  for (int l = 0; l < 64; ++l)
    for (int r = 0; r < 16; ++r) {
      int row = cv == 0 ? row_c0(r, l >> 5) : cv == 1 ? row_c1(r, l >> 5)
                                                      : row_c2(r, l >> 5);
      int col = l & 31;
      if (fabsf(raw[l * 16 + r] - ref[row * 32 + col]) > 0.5f) ++bad;
    }
  printf("%s C-map%d: %s (%d/1024 wrong)\n", tag, cv,
         bad == 0 ? "MATCH" : "no", bad);
  return bad == 0;
}

int main() {
  float hA[32 * 16], hB[16 * 32], hRef[32 * 32], hRaw[64 * 16];
  // small ints: exact in bf16 and f32
  for (int i = 0; i < 32 * 16; ++i) hA[i] = (float)((i * 7 + 3) % 17 - 8);
  for (int i = 0; i < 16 * 32; ++i) hB[i] = (float)((i * 5 + 1) % 15 - 7);
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j) {
      float s = 0;
      for (int k = 0; k < 16; ++k) s += hA[i * 16 + k] * hB[k * 32 + j];
      hRef[i * 32 + j] = s;
    }
  float *dA, *dB, *dRaw;
  hipMalloc(&dA, sizeof(hA));
  hipMalloc(&dB, sizeof(hB));
  hipMalloc(&dRaw, sizeof(hRaw));
  hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);

  // ---- probe 1: f32 (K=2) with documented A/B maps -> find C map ----
  float hA2[32 * 2], hB2[2 * 32], hRef2[32 * 32];
  for (int i = 0; i < 64; ++i) hA2[i] = (float)((i * 3 + 2) % 19 - 9);
  for (int i = 0; i < 64; ++i) hB2[i] = (float)((i * 11 + 5) % 13 - 6);
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j)
      hRef2[i * 32 + j] =
          hA2[i * 2] * hB2[j] + hA2[i * 2 + 1] * hB2[32 + j];
  float *dA2, *dB2;
  hipMalloc(&dA2, sizeof(hA2));
  hipMalloc(&dB2, sizeof(hB2));
  hipMemcpy(dA2, hA2, sizeof(hA2), hipMemcpyHostToDevice);
  hipMemcpy(dB2, hB2, sizeof(hB2), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe_f32, dim3(1), dim3(64), 0, 0, dA2, dB2, dRaw);
  hipMemcpy(hRaw, dRaw, sizeof(hRaw), hipMemcpyDeviceToHost);
  for (int cv = 0; cv < 3; ++cv) check(hRaw, hRef2, cv, "f32_32x32x2");

  // ---- probe 2: bf16 fill-map candidates under each C map ----
#define RUN(AV, BV)                                                       \
  do {                                                                    \
    hipLaunchKernelGGL((probe_bf16<AV, BV>), dim3(1), dim3(64), 0, 0, dA, \
                       dB, dRaw);                                         \
    hipMemcpy(hRaw, dRaw, sizeof(hRaw), hipMemcpyDeviceToHost);           \
    for (int cv = 0; cv < 3; ++cv)                                        \
      check(hRaw, hRef, cv, "bf16 A" #AV "B" #BV);                        \
  } while (0)
  RUN(0, 0);
  RUN(0, 1);
  RUN(1, 0);
  RUN(1, 1);
  // dump a corner of the raw acc for manual inspection if nothing matched
  printf("raw lane0 regs: ");
  for (int r = 0; r < 16; ++r) printf("%.0f ", hRaw[r]);
  printf("\nref row0: ");
  for (int j = 0; j < 8; ++j) printf("%.0f ", hRef[j]);
  printf("\nref col0 rows0-15: ");
  for (int i = 0; i < 16; ++i) printf("%.0f ", hRef[i * 32]);
  printf("\n");
  return 0;
}
