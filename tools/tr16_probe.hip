// tr16_probe.hip — empirical lane->element mapping of
// __builtin_amdgcn_ds_read_tr16_b64_v4i16 (v_ds_read_b64_tr_b16) on gfx950.
// LDS is filled with identity (lds[i] = i); each lane reads at a chosen
// base and we print which LDS indices landed in which lane/element.
//   hipcc --offload-arch=gfx950 tools/tr16_probe.hip -o /tmp/tr16 && /tmp/tr16
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) short short4v;

template <int MODE>
__global__ void probe(short* out) {
  __shared__ short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  const int lane = threadIdx.x;
  int base;
  if (MODE == 0) base = lane * 4;             // contiguous 8B per lane
  else if (MODE == 1) base = (lane / 16) * 64 + (lane % 16) * 4;
  else base = (lane % 16) * 64 + (lane / 16) * 4;  // row-major-ish tile
  auto p = (__attribute__((address_space(3))) short4v*)(&lds[base]);
  short4v v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = v[j];
}

int main() {
  short* d;
  hipMalloc(&d, 64 * 4 * 2);
  short h[256];
  const char* names[3] = {"lane*4 contiguous", "(l/16)*64+(l%16)*4",
                          "(l%16)*64+(l/16)*4"};
  for (int m = 0; m < 3; ++m) {
    if (m == 0) hipLaunchKernelGGL(probe<0>, dim3(1), dim3(64), 0, 0, d);
    if (m == 1) hipLaunchKernelGGL(probe<1>, dim3(1), dim3(64), 0, 0, d);
    if (m == 2) hipLaunchKernelGGL(probe<2>, dim3(1), dim3(64), 0, 0, d);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("== MODE %d: base = %s ==\n", m, names[m]);
    for (int l = 0; l < 64; ++l) {
      printf("lane %2d: %4d %4d %4d %4d\n", l, h[l * 4], h[l * 4 + 1],
             h[l * 4 + 2], h[l * 4 + 3]);
      if (l == 19 && m > 0) { printf("  ...\n"); l = 47; }
    }
  }
  return 0;
}
