// fp_standalone.hip — standalone harness for flash_prefill debugging.
// Reuses the production kernel source directly; K=0 / one-hot V probe on a
// minimal mixed-wave shape (chunk*qpg not a multiple of 32), then a random
// value check vs a CPU reference.
//   hipcc --offload-arch=gfx950 -DFP_DEBUG tools/fp_standalone.hip -o /tmp/fps
#include "../csrc/hip/flash_prefill.hip"

#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

static float b2f(short u) {
  union { float f; unsigned i; } c;
  c.i = ((unsigned)(unsigned short)u) << 16;
  return c.f;
}
static short f2b(float f) {
  union { float f; unsigned i; } c;
  c.f = f;
  unsigned lsb = (c.i >> 16) & 1;
  return (short)((c.i + 0x7FFF + lsb) >> 16);
}

int main(int argc, char** argv) {
  int chunk = argc > 1 ? atoi(argv[1]) : 40;
  int qpg = argc > 2 ? atoi(argv[2]) : 1;
  int prior = argc > 3 ? atoi(argv[3]) : 0;
  int kvh = argc > 4 ? atoi(argv[4]) : 1;
  int probe = argc > 5 ? atoi(argv[5]) : 1;   // 1 = K=0/one-hot V
  const int Dh = 128, BS = 16;
  int ctx = prior + chunk, qh = kvh * qpg, T = chunk;
  int nb = (ctx + BS - 1) / BS;
  int vrows = chunk * qpg;
  printf("chunk=%d qpg=%d prior=%d kvh=%d vrows=%d probe=%d\n", chunk, qpg,
         prior, kvh, vrows, probe);

  std::vector<short> hq((size_t)T * qh * Dh), hk((size_t)nb * kvh * BS * Dh),
      hv(hk.size()), hout(hq.size(), (short)0x7FC0 /*poison*/);
  srand(7);
  auto rnd = []() { return (float)(rand() % 2001 - 1000) / 500.f; };
  for (auto& x : hq) x = f2b(rnd());
  if (probe) {
    for (auto& x : hk) x = 0;
    for (auto& x : hv) x = 0;
    for (int t = 0; t < ctx; ++t)
      for (int h = 0; h < kvh; ++h)
        hv[(((size_t)(t / BS) * kvh + h) * BS + t % BS) * Dh + t % Dh] =
            f2b(1.f);
  } else {
    for (auto& x : hk) x = f2b(rnd());
    for (auto& x : hv) x = f2b(rnd());
  }
  std::vector<int32_t> hbt(nb), hmeta = {0, chunk, prior};
  for (int i = 0; i < nb; ++i) hbt[i] = i;
  std::vector<int32_t> htiles;
  for (int v0 = 0; v0 < vrows; v0 += 128) {
    htiles.push_back(0);
    htiles.push_back(v0);
  }
  short *dq, *dk, *dv, *dout;
  int32_t *dbt, *dmeta, *dtiles;
  hipMalloc(&dq, hq.size() * 2);
  hipMalloc(&dk, hk.size() * 2);
  hipMalloc(&dv, hv.size() * 2);
  hipMalloc(&dout, hout.size() * 2);
  hipMalloc(&dbt, hbt.size() * 4);
  hipMalloc(&dmeta, hmeta.size() * 4);
  hipMalloc(&dtiles, htiles.size() * 4);
  hipMemcpy(dq, hq.data(), hq.size() * 2, hipMemcpyHostToDevice);
  hipMemcpy(dk, hk.data(), hk.size() * 2, hipMemcpyHostToDevice);
  hipMemcpy(dv, hv.data(), hv.size() * 2, hipMemcpyHostToDevice);
  hipMemcpy(dout, hout.data(), hout.size() * 2, hipMemcpyHostToDevice);
  hipMemcpy(dbt, hbt.data(), hbt.size() * 4, hipMemcpyHostToDevice);
  hipMemcpy(dmeta, hmeta.data(), hmeta.size() * 4, hipMemcpyHostToDevice);
  hipMemcpy(dtiles, htiles.data(), htiles.size() * 4, hipMemcpyHostToDevice);
  hipError_t e = lds_flash_prefill(dq, dk, dv, dbt, dmeta, dtiles, dout,
                                   (int)htiles.size() / 2, qh, kvh, BS, Dh,
                                   nb, 1.f / sqrtf(128.f), 0);
  hipDeviceSynchronize();
  printf("launch: %s\n", hipGetErrorString(e));
  hipMemcpy(hout.data(), dout, hout.size() * 2, hipMemcpyDeviceToHost);

  // CPU reference
  int bad = 0, poison = 0;
  for (int p = 0; p < chunk; ++p)
    for (int h = 0; h < qh; ++h) {
      int kh = h / qpg;
      int nvalid = prior + p + 1;
      double m = -1e30;
      std::vector<double> s(nvalid);
      for (int t = 0; t < nvalid; ++t) {
        double dot = 0;
        for (int d = 0; d < Dh; ++d)
          dot += b2f(hq[((size_t)p * qh + h) * Dh + d]) *
                 b2f(hk[(((size_t)(t / BS) * kvh + kh) * BS + t % BS) * Dh +
                        d]);
        s[t] = dot / sqrt(128.0);
        if (s[t] > m) m = s[t];
      }
      double l = 0;
      for (int t = 0; t < nvalid; ++t) {
        s[t] = exp(s[t] - m);
        l += s[t];
      }
      double maxd = 0;
      int maxdim = -1;
      for (int d = 0; d < Dh; ++d) {
        double o = 0;
        for (int t = 0; t < nvalid; ++t)
          o += s[t] *
               b2f(hv[(((size_t)(t / BS) * kvh + kh) * BS + t % BS) * Dh + d]);
        o /= l;
        float got = b2f(hout[((size_t)p * qh + h) * Dh + d]);
        if ((unsigned short)hout[((size_t)p * qh + h) * Dh + d] == 0x7FC0)
          ++poison;
        if (fabs(got - o) > maxd) { maxd = fabs(got - o); maxdim = d; }
      }
      if (maxd > 0.03) {
        if (bad < 10)
          printf("  BAD p=%d h=%d maxdiff=%.4f at dim %d got=%f\n", p, h,
                 maxd, maxdim,
                 b2f(hout[((size_t)p * qh + h) * Dh + maxdim]));
        ++bad;
      }
    }
  printf("bad rows: %d / %d   poison shorts: %d\n", bad, chunk * qh, poison);
  return bad ? 1 : 0;
}
