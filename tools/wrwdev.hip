// Standalone dev driver for the wrw2 kernel (no torch): small problem,
// CPU reference, per-16x16-block error map. Build:
//   hipcc --offload-arch=gfx950 -O2 tools/wrwdev.hip csrc/conv_wrw2.hip \
//         -o tools/bin/wrwdev
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <cmath>

typedef unsigned short bf16_t;

extern "C" int tfosr_wrw2_split(int, int, long);
extern "C" void tfosr_conv_wrw2(const void*, const void*, const void*, float*,
                                float*, int, int, int, int, int, int, int,
                                int, int, int, int, int, hipStream_t);

static bf16_t f2b(float f) {
  unsigned u;
  __builtin_memcpy(&u, &f, 4);
  unsigned r = (u + 0x7FFF + ((u >> 16) & 1)) >> 16;
  return (bf16_t)r;
}
static float b2f(bf16_t b) {
  unsigned u = (unsigned)b << 16;
  float f;
  __builtin_memcpy(&f, &u, 4);
  return f;
}

int main(int argc, char** argv) {
  int N = 1, Cin = 64, H = 8, W = 8, Cout = 64, K = 1, stride = 1, P = 0;
  if (argc > 1) K = atoi(argv[1]);
  if (argc > 2) Cout = atoi(argv[2]);
  if (argc > 3) Cin = atoi(argv[3]);
  if (argc > 4) stride = atoi(argv[4]);
  if (argc > 5) { H = atoi(argv[5]); W = H; }
  if (argc > 6) N = atoi(argv[6]);
  if (argc > 7) W = atoi(argv[7]);
  if (K == 3) P = 1;
  int OH = (H + 2 * P - K) / stride + 1, OW = (W + 2 * P - K) / stride + 1;
  long M = (long)N * OH * OW;
  int taps = K * K;
  long dWn = (long)Cout * taps * Cin;

  std::vector<bf16_t> hx(N * H * W * Cin), hdy(M * Cout);
  srand(42);
  for (auto& v : hx) v = f2b((rand() % 200 - 100) / 50.f);
  for (auto& v : hdy) v = f2b((rand() % 200 - 100) / 50.f);

  // reference
  std::vector<float> ref(dWn, 0.f);
  for (int n = 0; n < N; ++n)
    for (int oh = 0; oh < OH; ++oh)
      for (int ow = 0; ow < OW; ++ow)
        for (int r = 0; r < K; ++r)
          for (int s = 0; s < K; ++s) {
            int ih = oh * stride - P + r, iw = ow * stride - P + s;
            if (ih < 0 || ih >= H || iw < 0 || iw >= W) continue;
            const bf16_t* xp = &hx[(((long)n * H + ih) * W + iw) * Cin];
            const bf16_t* dp = &hdy[(((long)n * OH + oh) * OW + ow) * Cout];
            for (int co = 0; co < Cout; ++co)
              for (int ci = 0; ci < Cin; ++ci)
                ref[(long)co * taps * Cin + (r * K + s) * Cin + ci] +=
                    b2f(dp[co]) * b2f(xp[ci]);
          }

  bf16_t *dx, *ddy, *dg;
  float *dws, *ddW;
  int split = tfosr_wrw2_split(Cout, Cin, M);
  (void)hipMalloc(&dx, hx.size() * 2);
  (void)hipMalloc(&ddy, hdy.size() * 2);
  (void)hipMalloc(&dg, 128);
  (void)hipMemset(dg, 0, 128);
  (void)hipMalloc(&dws, (long)split * dWn * 4);
  // poison the workspace so unwritten entries are visible
  (void)hipMemset(dws, 0xFF, (long)split * dWn * 4);
  (void)hipMalloc(&ddW, dWn * 4);
  (void)hipMemcpy(dx, hx.data(), hx.size() * 2, hipMemcpyHostToDevice);
  (void)hipMemcpy(ddy, hdy.data(), hdy.size() * 2, hipMemcpyHostToDevice);

  tfosr_conv_wrw2(ddy, dx, dg, dws, ddW, N, H, W, Cin, Cout, OH, OW, K, K,
                  stride, P, split, 0);
  hipError_t e = hipDeviceSynchronize();
  printf("split=%d M=%ld err=%s\n", split, M, hipGetErrorString(e));

  std::vector<float> out(dWn);
  (void)hipMemcpy(out.data(), ddW, dWn * 4, hipMemcpyDeviceToHost);
  double maxerr = 0;
  long nnan = 0;
  for (long i = 0; i < dWn; ++i) {
    if (std::isnan(out[i])) { ++nnan; continue; }
    double d = fabs(out[i] - ref[i]);
    if (d > maxerr) maxerr = d;
  }
  printf("maxerr=%.4f nan=%ld of %ld\n", maxerr, nnan, dWn);
  // block map (cout block x K-col block of 16)
  int nb = (int)(taps * Cin / 16), mb = Cout / 16;
  if (nb > 24) nb = 24;
  for (int bi = 0; bi < mb && bi < 8; ++bi) {
    printf("coutblk %d: ", bi);
    for (int bj = 0; bj < nb; ++bj) {
      double e2 = 0; bool nan2 = false;
      for (int i = 0; i < 16; ++i)
        for (int j = 0; j < 16; ++j) {
          float v = out[(long)(bi * 16 + i) * taps * Cin + bj * 16 + j];
          if (std::isnan(v)) nan2 = true;
          else {
            double d = fabs(v - ref[(long)(bi * 16 + i) * taps * Cin + bj * 16 + j]);
            if (d > e2) e2 = d;
          }
        }
      printf(nan2 ? "  NAN" : " %4.1f", e2);
    }
    printf("\n");
  }
  // first row sample
  printf("got : ");
  for (int j = 0; j < 8; ++j) printf("%7.2f", out[j]);
  printf("\nwant: ");
  for (int j = 0; j < 8; ++j) printf("%7.2f", ref[j]);
  printf("\n");
  return 0;
}
