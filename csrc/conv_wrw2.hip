// Weight-gradient (wrw) kernel for gfx950 — MFMA over the M (pixel) axis.
//
//   dW[cout][(r,s,cin)] = sum_m dy[m][cout] * x[tap(m,r,s)][cin]
//
// Both operands are stored M-major with channels fast (NHWC), but the MFMA
// contraction runs over M — every fragment is a *transposed* view. Round 1's
// attempt staged transposed via scalar ds_write_b16 and lost 3x to MIOpen
// (commit 845fb1f). This kernel instead:
//
//   * stages dy / per-tap x tiles with coalesced 16 B global_load_lds chunks
//     into plain row-major [32 m][16 c] blocks, then uses gfx950's hardware
//     transpose-read ds_read_b64_tr_b16 to pull k(=m)-major MFMA fragments.
//     Measured semantics (tools/trprobe.hip on MI355X): dest[l][j] =
//     lds[align8B(addr_of_lane((l&48) + 4j + ((l>>2)&3))) + (l&3) elems].
//     With per-lane address ((l>>4)*8 + 4t + ((l&15)>>2))*16 + (l&3)*4
//     elements into the row-major block, lane l receives exactly
//     A[channel = l&15][m = (l>>4)*8 + 4t + j] — the MFMA operand — while
//     staging stays 16 B contiguous (channels are the fast axis in HBM).
//   * split-M: each workgroup owns a contiguous m-range and writes its
//     partial tile to a workspace slice (plain stores); a tiny reduce kernel
//     sums the slices — no global atomics.
//
// Tiles: COUT_T=64 x CIN_T=64 x taps, 8 waves as a 2(cout)x4(cin) grid,
// K-step = 32 pixels, 3-deep LDS ring with counted vmcnt (never drains to 0
// in the main loop). One kernel covers 1x1 and 3x3, stride 1 and 2 (stride
// only changes the per-chunk x gather — chunks stay contiguous in cin).
#include "tfosr_common.h"

typedef short s16x4 __attribute__((ext_vector_type(4)));
typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define AS3 __attribute__((address_space(3)))

__device__ __forceinline__ void wrw_stage16(const char* src, AS3 char* dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)src, (AS3 void*)dst, 16,
      0, 0);
}

// magic-multiply division: q = (m * magic) >> 40, magic = 2^40/d + 1
__device__ __forceinline__ unsigned wrw_div(unsigned m, unsigned long magic) {
  return (unsigned)(((unsigned long)m * magic) >> 40);
}

// TAPS = R*S (1 or 9); regions = TAPS+1 (dy first), each 32m x 64c
template <int TAPS, int DEPTH>
__global__ __launch_bounds__(512, 1) void conv_wrw2_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ X,
    const bf16_t* __restrict__ guard, float* __restrict__ ws,
    int Nn, int H, int W, int Cin, int Cout, int OH, int OW, int FW,
    int stride, int P, int steps_per_wg, long M,
    unsigned long magicOW, unsigned long magicOHOW, int ntn, int split) {
  constexpr int REG = TAPS + 1;
  constexpr int REGB = 32 * 64 * 2;           // 4 KB per region
  constexpr int SLOT = REG * REGB;
  constexpr int CPT = REG * 256 / 512;        // chunks per thread per slot
  static_assert(REG * 256 % 512 == 0, "TAPS must be odd");
  __shared__ char lds[DEPTH * SLOT];
  AS3 char* lds3 = (AS3 char*)lds;

  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {  // XCD-aware linearization (8 XCDs round-robin by hardware)
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tile = wgid / split;       // output tile (cout x cin blocks)
  const int sp = wgid - tile * split;  // split-M slice
  const int cout0 = (tile / ntn) * 64;
  const int cin0 = (tile % ntn) * 64;

  const long m_begin = (long)sp * steps_per_wg * 32;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 2;            // 0..1: cout half (32 couts)
  const int wc = wave & 3;             // 0..3: cin block (16 cins)

  // per-thread chunk descriptors (fixed across k-steps except the m part)
  int ch_region[CPT], ch_mloc[CPT], ch_ldsoff[CPT];
  long ch_coff[CPT];   // channel byte offset into the global row; <0 => guard
  // global_load_lds writes lane-linearly from the wave base, so the LDS
  // destination is the plain linear chunk index; the (region, block, m, h)
  // the chunk HOLDS is decoded from that index instead.
  #pragma unroll
  for (int u = 0; u < CPT; ++u) {
    int q = t + u * 512;               // linear 16 B chunk index
    int region = q >> 8;
    int m = (q >> 1) & 31, cb = (q >> 6) & 3, h = q & 1;
    ch_region[u] = region;
    ch_mloc[u] = m;
    ch_ldsoff[u] = q * 16;
    int cglob = (region == 0 ? cout0 : cin0) + cb * 16 + h * 8;
    int climit = region == 0 ? Cout : Cin;
    ch_coff[u] = (cglob + 7 < climit) ? (long)cglob * 2 : -1;
  }

  auto stage = [&](int slot, long m0) {
    AS3 char* base = lds3 + slot * SLOT;
    #pragma unroll
    for (int u = 0; u < CPT; ++u) {
      const char* src = (const char*)guard + (t & 3) * 16;
      long m = m0 + ch_mloc[u];
      if (m < M && ch_coff[u] >= 0) {
        unsigned mm = (unsigned)m;
        unsigned n = wrw_div(mm, magicOHOW);
        unsigned rem = mm - n * (unsigned)(OH * OW);
        unsigned oh = wrw_div(rem, magicOW);
        unsigned ow = rem - oh * (unsigned)OW;
        if (ch_region[u] == 0) {
          src = (const char*)dy + (m * Cout) * 2 + ch_coff[u];
        } else {
          int tap = ch_region[u] - 1;
          int r = tap / FW, s = tap - (tap / FW) * FW;
          int ih = (int)oh * stride - P + r, iw = (int)ow * stride - P + s;
          if (ih >= 0 && ih < H && iw >= 0 && iw < W)
            src = (const char*)X +
                (((long)n * H + ih) * W + iw) * (long)Cin * 2 + ch_coff[u];
        }
      }
      wrw_stage16(src, base + ch_ldsoff[u]);
    }
  };

  // accumulators: 2 cout blocks x TAPS tiles of 16x16
  f32x4 acc[2][TAPS];
  #pragma unroll
  for (int i = 0; i < 2; ++i)
    #pragma unroll
    for (int j = 0; j < TAPS; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  long nkt = (M - m_begin + 31) / 32;
  if (nkt > steps_per_wg) nkt = steps_per_wg;
  if (nkt <= 0) nkt = 0;

  // per-lane transpose-read address into a row-major [32m][16c] block
  // (bytes); the t=1 half (m += 4) sits +128 B further
  const int lbase =
      ((((lane >> 4) * 8 + ((lane & 15) >> 2)) * 16) + (lane & 3) * 4) * 2;

  auto trread8 = [&](AS3 char* region, int blk) -> bf16x8 {
    AS3 s16x4* p =
        (AS3 s16x4*)(region + blk * 1024 + lbase);
    s16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
    s16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (AS3 s16x4*)((AS3 char*)p + 128));
    bf16x8 f;
    f[0] = lo[0]; f[1] = lo[1]; f[2] = lo[2]; f[3] = lo[3];
    f[4] = hi[0]; f[5] = hi[1]; f[6] = hi[2]; f[7] = hi[3];
    return f;
  };

  #pragma unroll
  for (int pf = 0; pf < DEPTH - 1; ++pf)
    if (pf < nkt) stage(pf, m_begin + (long)pf * 32);
  // after staging slot kt+DEPTH-1, at most (DEPTH-1) newer slots in flight
  constexpr int INFLIGHT = (DEPTH - 1) * CPT;

  for (long kt = 0; kt < nkt; ++kt) {
    if (kt + DEPTH - 1 < nkt)
      stage((int)((kt + DEPTH - 1) % DEPTH), m_begin + (kt + DEPTH - 1) * 32);
    asm volatile("s_waitcnt vmcnt(%0)" ::"i"(INFLIGHT) : "memory");
    __builtin_amdgcn_sched_barrier(0);
    __syncthreads();

    AS3 char* slot = lds3 + (kt % DEPTH) * SLOT;
    bf16x8 afrag[2];
    #pragma unroll
    for (int co = 0; co < 2; ++co)
      afrag[co] = trread8(slot, wr * 2 + co);
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int tap = 0; tap < TAPS; ++tap) {
      bf16x8 b = trread8(slot + (tap + 1) * REGB, wc);
      #pragma unroll
      for (int co = 0; co < 2; ++co)
        acc[co][tap] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[co], b, acc[co][tap], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

  // write the partial tile to workspace slice sp (plain fp32 stores)
  const long K = (long)TAPS * Cin;
  float* out = ws + (long)sp * Cout * K;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  const int cin_g = cin0 + wc * 16 + ccol;
  if (cin_g < Cin) {
    #pragma unroll
    for (int co = 0; co < 2; ++co) {
      int cout_g0 = cout0 + wr * 32 + co * 16 + crow0;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int cout_g = cout_g0 + r;
        if (cout_g >= Cout) continue;
        #pragma unroll
        for (int tap = 0; tap < TAPS; ++tap)
          out[(long)cout_g * K + (long)tap * Cin + cin_g] = acc[co][tap][r];
      }
    }
  }
}

// dW[i] = sum over split slices of ws[s][i]
__global__ void wrw_reduce_kernel(const float* __restrict__ ws,
                                  float* __restrict__ dW, long n, int split) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  float acc = 0.f;
  for (int s = 0; s < split; ++s) acc += ws[(long)s * n + i];
  dW[i] = acc;
}

extern "C" {

// returns required split count for the given problem (grid sizing guidance)
int tfosr_wrw2_split(int Cout, int Cin, long M) {
  int ntiles = ((Cout + 63) / 64) * ((Cin + 63) / 64);
  int split = 1024 / ntiles;          // target >=1024 workgroups
  if (split < 1) split = 1;
  long steps = (M + 31) / 32;
  while ((long)split * 8 > steps && split > 1) split /= 2;
  return split;
}

void tfosr_conv_wrw2(const void* dy, const void* x, const void* guard,
                     float* ws, float* dW, int N, int H, int W, int Cin,
                     int Cout, int OH, int OW, int R, int S_f, int stride,
                     int P, int split, hipStream_t s) {
  const long M = (long)N * OH * OW;
  const int taps = R * S_f;
  const unsigned long magicOW = (1UL << 40) / (unsigned)OW + 1;
  const unsigned long magicOHOW = (1UL << 40) / (unsigned)(OH * OW) + 1;
  const int ntn = (Cin + 63) / 64;
  const int ntiles = ((Cout + 63) / 64) * ntn;
  const long steps = (M + 31) / 32;
  const int spw = (int)((steps + split - 1) / split);
  dim3 grid(ntiles * split), block(512);
  if (taps == 9)
    hipLaunchKernelGGL((conv_wrw2_kernel<9, 3>), grid, block, 0, s,
                       (const bf16_t*)dy, (const bf16_t*)x,
                       (const bf16_t*)guard, ws, N, H, W, Cin, Cout, OH, OW,
                       S_f, stride, P, spw, M, magicOW, magicOHOW, ntn, split);
  else if (taps == 1)
    hipLaunchKernelGGL((conv_wrw2_kernel<1, 3>), grid, block, 0, s,
                       (const bf16_t*)dy, (const bf16_t*)x,
                       (const bf16_t*)guard, ws, N, H, W, Cin, Cout, OH, OW,
                       S_f, stride, P, spw, M, magicOW, magicOHOW, ntn, split);
  const long n = (long)Cout * taps * Cin;
  dim3 rg((n + 255) / 256), rb(256);
  hipLaunchKernelGGL(wrw_reduce_kernel, rg, rb, 0, s, ws, dW, n, split);
}

}  // extern "C"
