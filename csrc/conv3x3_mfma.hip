// Implicit-GEMM 3x3 convolution for gfx950 (NHWC/channels_last, bf16):
//
//   out[M = N*OH*OW, Cout] = im2col(x)[M, K=9*Cin] @ W9[Cout, K]^T
//
// W9 is the weight pre-permuted to [Cout][r][s][Cin] so both operands are
// K-contiguous, exactly the gemm_bt layout. The A matrix is *virtual*: the
// staging stage computes each 16 B chunk's global source from (m -> n,oh,ow)
// and the K-step's uniform (r, s, cin0) — valid because Cin % 32 == 0 makes
// every 32-wide K-step live inside a single (r,s) run. Out-of-image taps
// (padding) redirect the global_load_lds source to a 64 B zero guard buffer.
//
// Same 4-deep LDS ring + counted-vmcnt pipeline as gemm_bt256 (no vmcnt(0)
// in the main loop); two tile shapes: 256x256 (Cout >= 192) and 256x128.
// Backward-data for stride 1 reuses this kernel: dx = conv3x3(dy, W') with
// W'[cin][r][s][cout] = W[cout][2-r][2-s][cin] (built host-side).
//
// Input dilation D (template): the input is read as if zero-dilated by D —
// virtual index ihv maps to stored row ihv/D and contributes only when
// ihv % D == 0 (otherwise the tap redirects to the zero guard). This makes
// the SAME kernel compute:
//   - stride-2 forward            (S=2, D=1)
//   - stride-2 backward-data      (S=1, D=2, W' flipped: dx = conv(dy_dil, W'))
//   - ConvTranspose2d k3 s2       (S=1, D=2, W swapped/flipped)
// and, with K = 1*Cin (w shaped [Cout, Cin], taps degenerate to r=s=0),
// 1x1 convolutions at any stride plus their dilated backward-data.
#include "tfosr_common.h"

typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define C3_BK 32
#define C3_SWZ(l) ((l) ^ ((((l) >> 7) & 7) << 4))

__device__ __forceinline__ void c3_stage16(const char* src,
                                           __attribute__((address_space(3))) char* dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)src,
      (__attribute__((address_space(3))) void*)dst, 16, 0, 0);
}

// WRG x WCG wave grid (8 waves); per-wave output (MI*16) x 64
//
// PAR mode (parity-decomposed backward-data / transposed conv): the grid
// covers ONE output parity class — output pixel (oh*2+oh0, ow*2+ow0) over
// subsampled dims OH x OW — with the class's tap subset supplied as a packed
// (r<<4|s) byte list in tap_pack. Every tap of a class hits stored input
// rows exactly (the D=2 divisibility always holds), so none of the 4x
// dilation-zero MFMA waste of the plain D=2 formulation remains. Stores
// land at the class's strided pixels of the REAL output (OWr/total given).
template <typename OT, int WRG, int WCG, int MI, int D = 1, bool ACC = false,
          bool PAR = false>
__global__ __launch_bounds__(512, 1) void conv3x3_kernel(
    const bf16_t* __restrict__ X, const bf16_t* __restrict__ W9,
    const bf16_t* __restrict__ guard, OT* __restrict__ C,
    int Nn, int H, int Wd, int Cin, int Cout, int OH, int OW, int S, int P,
    int taps, int fw, int pixst, unsigned long tap_pack, int oh0, int ow0,
    int OWr, long OHOWr) {
  constexpr int BM = WRG * MI * 16;
  constexpr int BN = WCG * 64;
  constexpr int ABYTES = BM * C3_BK * 2;
  constexpr int BBYTES = BN * C3_BK * 2;
  constexpr int ACHUNK = ABYTES / 16 / 512;
  constexpr int BCHUNK = BBYTES / 16 / 512;
  constexpr int SLOT = ABYTES + BBYTES;
  __shared__ char lds[4 * SLOT];
  __attribute__((address_space(3))) char* lds3 =
      (__attribute__((address_space(3))) char*)lds;

  const int K = taps * Cin;
  const long M = (long)Nn * OH * OW;
  const int ntn = (Cout + BN - 1) / BN;
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const long tile_m = (long)(wgid / ntn) * BM;
  const long tile_n = (long)(wgid % ntn) * BN;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave / WCG, wc = wave % WCG;

  // A chunks: output-pixel coords (uniform per thread across K-steps)
  int a_ohS[ACHUNK], a_owS[ACHUNK];  // oh*S - P, ow*S - P
  long a_pix[ACHUNK];                // (n*H*W + ...) base without (ih, iw)
  int a_col[ACHUNK];
  long a_n[ACHUNK];
  #pragma unroll
  for (int u = 0; u < ACHUNK; ++u) {
    int d = (t + u * 512) * 16;
    int sl = C3_SWZ(d);
    long row = sl >> 6;
    long m = tile_m + row;
    bool ok = m < M;
    if (!ok) m = 0;
    long n = m / ((long)OH * OW);
    long rem = m - n * (long)OH * OW;
    int oh = (int)(rem / OW), ow = (int)(rem - (long)(rem / OW) * OW);
    a_n[u] = ok ? n : -1;
    a_ohS[u] = (PAR ? oh * 2 + oh0 : oh) * S - P;
    a_owS[u] = (PAR ? ow * 2 + ow0 : ow) * S - P;
    a_pix[u] = n * (long)H * Wd;
    a_col[u] = sl & 63;
  }
  long srcB[BCHUNK];
  const long Kb = (long)K * 2;
  #pragma unroll
  for (int u = 0; u < BCHUNK; ++u) {
    int d = (t + u * 512) * 16;
    int sl = C3_SWZ(d);
    long row = sl >> 6;
    srcB[u] = ((tile_n + row < Cout) ? (tile_n + row) : (Cout - 1)) * Kb
              + (sl & 63);
  }

  auto stage_tile = [&](int slot, int kt) {
    const int k0 = kt * C3_BK;
    const int rs = k0 / Cin;
    const int rr = PAR ? (int)((tap_pack >> (rs * 8 + 4)) & 15)
                       : rs / fw;
    const int ss = PAR ? (int)((tap_pack >> (rs * 8)) & 15)
                       : rs - (rs / fw) * fw;
    const int cin0 = k0 - rs * Cin;
    __attribute__((address_space(3))) char* la = lds3 + slot * SLOT;
    __attribute__((address_space(3))) char* lb = la + ABYTES;
    #pragma unroll
    for (int u = 0; u < ACHUNK; ++u) {
      int ih = a_ohS[u] + rr, iw = a_owS[u] + ss;
      bool ok = a_n[u] >= 0 && ih >= 0 && iw >= 0;
      if (D > 1) {  // dilated input: only multiples of D are stored rows/cols
        ok = ok && (ih % D == 0) && (iw % D == 0);
        ih /= D;
        iw /= D;
      }
      const char* src;
      if (ok && ih < H && iw < Wd) {
        long pix = a_pix[u] + (long)ih * Wd + iw;
        src = (const char*)X + (pix * (long)pixst + cin0) * 2 + a_col[u];
      } else {
        src = (const char*)guard + a_col[u];
      }
      c3_stage16(src, la + (t + u * 512) * 16);
    }
    const long kbyte = (long)k0 * 2;
    #pragma unroll
    for (int u = 0; u < BCHUNK; ++u)
      c3_stage16((const char*)W9 + srcB[u] + kbyte, lb + (t + u * 512) * 16);
  };

  f32x4 acc[MI][4];
  #pragma unroll
  for (int i = 0; i < MI; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nkt = K / C3_BK;
  stage_tile(0, 0);
  if (1 < nkt) stage_tile(1, 1);
  if (2 < nkt) stage_tile(2, 2);

  const int frow = lane & 15;
  const int kslot = lane >> 4;
  constexpr int INFLIGHT = 2 * (ACHUNK + BCHUNK);  // 2 staged tiles in flight

  for (int kt = 0; kt < nkt; ++kt) {
    if (kt + 2 < nkt)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(INFLIGHT) : "memory");
    else  // tail: fewer newer slots in flight than the counted wait assumes
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __syncthreads();
    if (kt + 3 < nkt) stage_tile((kt + 3) & 3, kt + 3);

    __attribute__((address_space(3))) char* la = lds3 + (kt & 3) * SLOT;
    __attribute__((address_space(3))) char* lb = la + ABYTES;

    bf16x8 afrag[MI], bfrag[4];
    #pragma unroll
    for (int nj = 0; nj < 4; ++nj) {
      int l = (wc * 64 + nj * 16 + frow) * 64 + kslot * 16;
      bfrag[nj] = *(__attribute__((address_space(3))) bf16x8*)(lb + C3_SWZ(l));
    }
    #pragma unroll
    for (int mi = 0; mi < MI; ++mi) {
      int l = (wr * MI * 16 + mi * 16 + frow) * 64 + kslot * 16;
      afrag[mi] = *(__attribute__((address_space(3))) bf16x8*)(la + C3_SWZ(l));
    }
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int mi = 0; mi < MI; ++mi)
      #pragma unroll
      for (int nj = 0; nj < 4; ++nj)
        acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[mi], bfrag[nj], acc[mi][nj], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  #pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      long m = tile_m + wr * MI * 16 + mi * 16 + crow0 + r;
      if (m >= M) continue;
      if (PAR) {  // scatter to this parity class's pixels of the real output
        long n2 = m / ((long)OH * OW);
        long rem = m - n2 * (long)OH * OW;
        int oh = (int)(rem / OW), ow = (int)(rem - (long)(rem / OW) * OW);
        m = n2 * OHOWr + (long)(oh * 2 + oh0) * OWr + (ow * 2 + ow0);
      }
      #pragma unroll
      for (int nj = 0; nj < 4; ++nj) {
        long n = tile_n + wc * 64 + nj * 16 + ccol;
        if (n < Cout)
          C[m * Cout + n] = ACC
              ? (OT)((float)C[m * Cout + n] + acc[mi][nj][r])
              : (OT)acc[mi][nj][r];
      }
    }
  }
}

template <int D, bool ACC>
static void launch_conv(const void* X, const void* W9, const void* guard,
                        void* Y, int out_bf16, int N, int H, int W, int Cin,
                        int Cout, int OH, int OW, int S, int P, int taps,
                        int fw, int pixst, hipStream_t s) {
  const long M = (long)N * OH * OW;
  if (Cout >= 192) {
    int ntm = (int)((M + 255) / 256), ntn = (Cout + 255) / 256;
    dim3 grid(ntm * ntn);
    if (out_bf16)
      hipLaunchKernelGGL((conv3x3_kernel<bf16_t, 2, 4, 8, D, ACC>), grid,
                         dim3(512), 0, s, (const bf16_t*)X, (const bf16_t*)W9,
                         (const bf16_t*)guard, (bf16_t*)Y, N, H, W, Cin, Cout,
                         OH, OW, S, P, taps, fw, pixst, 0UL, 0, 0, OW, (long)OH * OW);
    else
      hipLaunchKernelGGL((conv3x3_kernel<float, 2, 4, 8, D, ACC>), grid,
                         dim3(512), 0, s, (const bf16_t*)X, (const bf16_t*)W9,
                         (const bf16_t*)guard, (float*)Y, N, H, W, Cin, Cout,
                         OH, OW, S, P, taps, fw, pixst, 0UL, 0, 0, OW, (long)OH * OW);
    return;
  }
  int ntm = (int)((M + 255) / 256), ntn = (Cout + 127) / 128;
  dim3 grid(ntm * ntn);
  if (out_bf16)
    hipLaunchKernelGGL((conv3x3_kernel<bf16_t, 4, 2, 4, D, ACC>), grid,
                       dim3(512), 0, s, (const bf16_t*)X, (const bf16_t*)W9,
                       (const bf16_t*)guard, (bf16_t*)Y, N, H, W, Cin, Cout,
                       OH, OW, S, P, taps, fw, pixst, 0UL, 0, 0, OW, (long)OH * OW);
  else
    hipLaunchKernelGGL((conv3x3_kernel<float, 4, 2, 4, D, ACC>), grid,
                       dim3(512), 0, s, (const bf16_t*)X, (const bf16_t*)W9,
                       (const bf16_t*)guard, (float*)Y, N, H, W, Cin, Cout,
                       OH, OW, S, P, taps, fw, pixst, 0UL, 0, 0, OW, (long)OH * OW);
}

extern "C" {

void tfosr_conv3x3(const void* X, const void* W9, const void* guard, void* Y,
                   int out_bf16, int N, int H, int W, int Cin, int Cout,
                   int OH, int OW, int S, int P, hipStream_t s) {
  launch_conv<1, false>(X, W9, guard, Y, out_bf16, N, H, W, Cin, Cout, OH, OW,
                        S, P, 9, 3, Cin, s);
}

// General entry: taps = fh*fw; input dilation Dil in {1, 2}
void tfosr_conv_mfma(const void* X, const void* W9, const void* guard, void* Y,
                     int out_bf16, int N, int H, int W, int Cin, int Cout,
                     int OH, int OW, int S, int P, int taps, int fw, int Dil,
                     int accum, hipStream_t s) {
  if (Dil == 2) {
    if (accum)
      launch_conv<2, true>(X, W9, guard, Y, out_bf16, N, H, W, Cin, Cout, OH,
                           OW, S, P, taps, fw, Cin, s);
    else
      launch_conv<2, false>(X, W9, guard, Y, out_bf16, N, H, W, Cin, Cout, OH,
                            OW, S, P, taps, fw, Cin, s);
  } else {
    if (accum)
      launch_conv<1, true>(X, W9, guard, Y, out_bf16, N, H, W, Cin, Cout, OH,
                           OW, S, P, taps, fw, Cin, s);
    else
      launch_conv<1, false>(X, W9, guard, Y, out_bf16, N, H, W, Cin, Cout, OH,
                            OW, S, P, taps, fw, Cin, s);
  }
}

// Parity-decomposed stride-2 backward-data / transposed-conv class launch:
// computes the (oh0, ow0) output parity class over subsampled dims OHs x OWs
// with ntaps class taps packed in tap_pack; writes strided pixels of the
// real output [.., OHr, OWr] (classes are disjoint; no accumulation needed).
void tfosr_conv_par(const void* X, const void* Wk, const void* guard, void* Y,
                    int N, int H, int W, int Cin, int Cout, int OHs, int OWs,
                    int P, int ntaps, unsigned long tap_pack, int oh0, int ow0,
                    int OWr, long OHOWr, int accum, hipStream_t s) {
  const long M = (long)N * OHs * OWs;
  if (Cout >= 192) {
    int ntm = (int)((M + 255) / 256), ntn = (Cout + 255) / 256;
    dim3 grid(ntm * ntn);
    if (accum)
      hipLaunchKernelGGL((conv3x3_kernel<bf16_t, 2, 4, 8, 2, true, true>),
                         grid, dim3(512), 0, s, (const bf16_t*)X,
                         (const bf16_t*)Wk, (const bf16_t*)guard, (bf16_t*)Y,
                         N, H, W, Cin, Cout, OHs, OWs, 1, P, ntaps, 1, Cin,
                         tap_pack, oh0, ow0, OWr, OHOWr);
    else
      hipLaunchKernelGGL((conv3x3_kernel<bf16_t, 2, 4, 8, 2, false, true>),
                         grid, dim3(512), 0, s, (const bf16_t*)X,
                         (const bf16_t*)Wk, (const bf16_t*)guard, (bf16_t*)Y,
                         N, H, W, Cin, Cout, OHs, OWs, 1, P, ntaps, 1, Cin,
                         tap_pack, oh0, ow0, OWr, OHOWr);
    return;
  }
  int ntm = (int)((M + 255) / 256), ntn = (Cout + 127) / 128;
  dim3 grid(ntm * ntn);
  if (accum)
    hipLaunchKernelGGL((conv3x3_kernel<bf16_t, 4, 2, 4, 2, true, true>), grid,
                       dim3(512), 0, s, (const bf16_t*)X, (const bf16_t*)Wk,
                       (const bf16_t*)guard, (bf16_t*)Y, N, H, W, Cin, Cout,
                       OHs, OWs, 1, P, ntaps, 1, Cin, tap_pack, oh0, ow0,
                       OWr, OHOWr);
  else
    hipLaunchKernelGGL((conv3x3_kernel<bf16_t, 4, 2, 4, 2, false, true>), grid,
                       dim3(512), 0, s, (const bf16_t*)X, (const bf16_t*)Wk,
                       (const bf16_t*)guard, (bf16_t*)Y, N, H, W, Cin, Cout,
                       OHs, OWs, 1, P, ntaps, 1, Cin, tap_pack, oh0, ow0,
                       OWr, OHOWr);
}

// ResNet stem 7x7/s2 forward over a pre-padded NHWC4 image (x4 layout:
// [N][230][230][4] bf16, Cin 3 zero-padded to 4, spatial pad 3 baked in).
// Each filter ROW is one uniform 32-wide K-step: "channels" = 8 px x 4 ch
// (7 real px + 1 whose weight columns are zero), pixel stride 4 elements.
// K = 7 x 32 = 224 vs the exact 147 — 1.52x MFMA inflation, far below the
// 10.7x a Cin-pad-to-32 implicit GEMM would cost, and no guard loads at all.
void tfosr_conv_stem(const void* X4, const void* W224, const void* guard,
                     void* Y, int N, int Hp, int Wp, int Cout, int OH, int OW,
                     hipStream_t s) {
  // geometry mapped onto the generic kernel: taps=7 rows (fw=1 so rr=rs),
  // S=2, P=0 (pre-padded), "Cin"=32, pixstride=4
  launch_conv<1, false>(X4, W224, guard, Y, /*out_bf16=*/1, N, Hp, Wp, 32,
                        Cout, OH, OW, 2, 0, 7, 1, 4, s);
}

}  // extern "C"
