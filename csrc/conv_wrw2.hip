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
// Tiles: COUT_T=64 x CIN_T=64 x taps, 4 waves each owning the full 64 couts
// x one 16-cin block (so every B fragment feeds 4 MFMAs and every A fragment
// 9 — 0.72 tr-reads per MFMA), K-step = 32 pixels, 3-deep LDS ring with
// counted vmcnt (never drains to 0 in the main loop). One kernel covers 1x1
// and 3x3, stride 1 and 2 (stride only changes the per-chunk x gather —
// chunks stay contiguous in cin).
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
__global__ __launch_bounds__(256, 1) void conv_wrw2_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ X,
    const bf16_t* __restrict__ guard, float* __restrict__ ws,
    int Nn, int H, int W, int Cin, int Cout, int OH, int OW, int FW,
    int stride, int P, int pixst, int steps_per_wg, long M,
    unsigned long magicOW, unsigned long magicOHOW, int ntn, int split) {
  constexpr int REG = TAPS + 1;
  constexpr int REGB = 32 * 64 * 2;           // 4 KB per region
  constexpr int SLOT = REG * REGB;
  constexpr int CPT = REG * 256 / 256;        // chunks per thread per slot
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
  const int wc = t >> 6;               // wave = 0..3: its 16-cin block

  // global_load_lds writes lane-linearly from the wave base, so the LDS
  // destination is the plain linear chunk index; the (region, block, m, h)
  // the chunk HOLDS is decoded from that index instead. Every chunk of a
  // thread holds the SAME pixel m ((t>>1)&31 — 256 = 0 mod 64), so the pixel
  // decode (magic divisions) hoists to once per thread per k-step: VALUBusy
  // measured 42% before this hoist, the dominant pipe.
  const int mloc = (t >> 1) & 31;
  int ch_isdy[CPT], ch_r[CPT], ch_s[CPT], ch_ldsoff[CPT];
  long ch_coff[CPT];
  #pragma unroll
  for (int u = 0; u < CPT; ++u) {
    int q = t + u * 256;               // linear 16 B chunk index
    int region = q >> 8;
    // bank stagger: the 8-channel half stored at physical slot (q&1) is the
    // logical half flipped by bit3 of m — otherwise the four kslot groups of
    // a transpose-read land on identical banks (4-way conflict, measured
    // ratio 2.0 in SQ_LDS_BANK_CONFLICT)
    int cb = (q >> 6) & 3, h = (q & 1) ^ ((q >> 4) & 1);
    int tap = region - 1;
    ch_isdy[u] = region == 0;
    ch_r[u] = region ? tap / FW - P : 0;
    ch_s[u] = region ? tap % FW - P : 0;
    ch_ldsoff[u] = q * 16;
    int cglob = (region == 0 ? cout0 : cin0) + cb * 16 + h * 8;
    int climit = region == 0 ? Cout : Cin;
    ch_coff[u] = (cglob + 7 < climit) ? (long)cglob * 2 : -1;
  }

  // Staging loads ALWAYS read a clamped in-bounds address with full EXEC:
  // global_load_lds takes its LDS base from M0 = readfirstlane(dst), so a
  // divergent (partial-EXEC) load takes M0 from whichever lane is first
  // active and every lane's write lands shifted (measured: tail pixels
  // double-counted). Invalid chunks (pad taps / m >= M) are recorded in a
  // per-slot bitmask and zeroed with per-lane ds_writes at consume time.
  unsigned invmask[DEPTH];
  auto stage = [&](int slot, long m0) {
    AS3 char* base = lds3 + slot * SLOT;
    long m = m0 + mloc;
    const bool mok = m < M;
    unsigned mm = (unsigned)(mok ? m : 0);
    unsigned n = wrw_div(mm, magicOHOW);
    unsigned rem = mm - n * (unsigned)(OH * OW);
    unsigned oh = wrw_div(rem, magicOW);
    unsigned ow = rem - oh * (unsigned)OW;
    const int ohS = (int)oh * stride, owS = (int)ow * stride;
    const char* dyrow = (const char*)dy + mm * (long)Cout * 2;
    const char* xbase = (const char*)X + (long)n * H * W * (long)pixst * 2;
    const long rowb = (long)W * pixst * 2, colb = (long)pixst * 2;
    unsigned inv = 0;
    #pragma unroll
    for (int u = 0; u < CPT; ++u) {
      int ih = ohS + ch_r[u], iw = owS + ch_s[u];
      bool inimg = ih >= 0 && ih < H && iw >= 0 && iw < W;
      int ihc = inimg ? ih : 0, iwc = inimg ? iw : 0;
      long coff = ch_coff[u] >= 0 ? ch_coff[u] : 0;
      const char* real = ch_isdy[u] ? dyrow + coff
                                    : xbase + ihc * rowb + iwc * colb + coff;
      if (!(mok && ch_coff[u] >= 0 && (ch_isdy[u] || inimg)))
        inv |= 1u << u;
      wrw_stage16(real, base + ch_ldsoff[u]);
    }
    invmask[slot] = inv;
  };
  auto zero_invalid = [&](int slot) {
    unsigned inv = invmask[slot];
    if (inv) {
      AS3 char* zb = lds3 + slot * SLOT;
      #pragma unroll
      for (int u = 0; u < CPT; ++u)
        if (inv & (1u << u)) {
          typedef int i32x4 __attribute__((ext_vector_type(4)));
          *(AS3 i32x4*)(zb + ch_ldsoff[u]) = (i32x4){0, 0, 0, 0};
        }
    }
  };

  // accumulators: 4 cout blocks x TAPS tiles of 16x16 (144 VGPRs at TAPS=9)
  f32x4 acc[4][TAPS];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < TAPS; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  long nkt = (M - m_begin + 31) / 32;
  if (nkt > steps_per_wg) nkt = steps_per_wg;
  if (nkt <= 0) nkt = 0;

  // per-lane transpose-read address into a row-major [32m][16c] block
  // (bytes); the t=1 half (m += 4) sits +128 B further
  const int lbase =
      ((((lane >> 4) * 8 + ((lane & 15) >> 2)) * 16) +
       ((lane & 3) ^ (((lane >> 4) & 1) << 1)) * 4) * 2;

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
    if (kt + DEPTH - 1 < nkt) {
      stage((int)((kt + DEPTH - 1) % DEPTH), m_begin + (kt + DEPTH - 1) * 32);
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(INFLIGHT) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    zero_invalid((int)(kt % DEPTH));
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __syncthreads();

    AS3 char* slot = lds3 + (kt % DEPTH) * SLOT;
    // issue every fragment read up front so the MFMA chain never serially
    // waits on LDS latency (1-2 waves/SIMD can't hide it otherwise)
    bf16x8 afrag[4], bfrag[TAPS];
    #pragma unroll
    for (int co = 0; co < 4; ++co)
      afrag[co] = trread8(slot, co);
    #pragma unroll
    for (int tap = 0; tap < TAPS; ++tap)
      bfrag[tap] = trread8(slot + (tap + 1) * REGB, wc);
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int tap = 0; tap < TAPS; ++tap)
      #pragma unroll
      for (int co = 0; co < 4; ++co)
        acc[co][tap] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[co], bfrag[tap], acc[co][tap], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    // drain LDS reads before the barrier: the compiler may sink the mfma
    // consumers (and their lgkm waits) past __syncthreads, leaving tr-reads
    // in flight while the next iteration's staging overwrites the slot
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
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
    for (int co = 0; co < 4; ++co) {
      int cout_g0 = cout0 + co * 16 + crow0;
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


// 1x1 wrw = plain TN GEMM over pixels: C[Cout, Cin] += dy^T @ x(strided).
// Tiles: COUT_T=128 x CIN_T=128, 4 waves each owning a 64x64 quadrant
// (1 tr-read per MFMA), 64-pixel step per barrier, 2-deep ring (64 KB ->
// 2 workgroups per CU so LDS latency hides across waves).
template <int DEPTH>
__global__ __launch_bounds__(256, 1) void wrw_gemm_tn_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ X,
    const bf16_t* __restrict__ guard, float* __restrict__ ws,
    int Nn, int H, int W, int Cin, int Cout, int OH, int OW,
    int stride, int steps_per_wg, long M,
    unsigned long magicOW, unsigned long magicOHOW, int ntn, int split) {
  constexpr int REGB = 64 * 128 * 2;          // 16 KB per region (64 m rows)
  constexpr int SLOT = 2 * REGB;
  constexpr int CPT = 8;                      // 2048 chunks / 256 threads
  __shared__ char lds[DEPTH * SLOT];
  AS3 char* lds3 = (AS3 char*)lds;

  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tile = wgid / split;
  const int sp = wgid - tile * split;
  const int cout0 = (tile / ntn) * 128;
  const int cin0 = (tile % ntn) * 128;

  const long m_begin = (long)sp * steps_per_wg * 64;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;    // 64-cout / 64-cin quadrant

  // each thread's chunks all hold pixel m = (t&127)>>1 (1024 chunks/region,
  // 256 threads): hoist the pixel decode to once per thread per k-step
  const int mloc = (t & 127) >> 1;
  int ch_isdy[CPT], ch_ldsoff[CPT];
  long ch_coff[CPT];
  #pragma unroll
  for (int u = 0; u < CPT; ++u) {
    int q = t + u * 256;
    int region = q >> 10;                     // 1024 chunks per region
    int qq = q & 1023;                        // within region: [64m][8 chunks]
    int cb = qq >> 7;                         // 8 channel blocks of 16
    int r2 = qq & 127;
    int h = (r2 & 1) ^ ((r2 >> 4) & 1);
    ch_isdy[u] = region == 0;
    ch_ldsoff[u] = q * 16;
    int cglob = (region == 0 ? cout0 : cin0) + cb * 16 + h * 8;
    int climit = region == 0 ? Cout : Cin;
    ch_coff[u] = (cglob + 7 < climit) ? (long)cglob * 2 : -1;
  }

  // full-EXEC clamped staging + consume-time zeroing of invalid chunks —
  // see conv_wrw2_kernel::stage for the divergent global_load_lds/M0 hazard
  unsigned invmask[DEPTH];
  auto stage = [&](int slot, long m0) {
    AS3 char* base = lds3 + slot * SLOT;
    long m = m0 + mloc;
    const bool mok = m < M;
    const long mm = mok ? m : 0;
    const char* dyrow = (const char*)dy + mm * (long)Cout * 2;
    const char* xrow;
    if (stride == 1) {
      xrow = (const char*)X + mm * (long)Cin * 2;
    } else {
      unsigned n = wrw_div((unsigned)mm, magicOHOW);
      unsigned rem = (unsigned)mm - n * (unsigned)(OH * OW);
      unsigned oh = wrw_div(rem, magicOW);
      unsigned ow = rem - oh * (unsigned)OW;
      xrow = (const char*)X +
          (((long)n * H + oh * stride) * W + ow * stride) * (long)Cin * 2;
    }
    unsigned inv = 0;
    #pragma unroll
    for (int u = 0; u < CPT; ++u) {
      long coff = ch_coff[u] >= 0 ? ch_coff[u] : 0;
      const char* real = (ch_isdy[u] ? dyrow : xrow) + coff;
      if (!(mok && ch_coff[u] >= 0)) inv |= 1u << u;
      wrw_stage16(real, base + ch_ldsoff[u]);
    }
    invmask[slot] = inv;
  };
  auto zero_invalid = [&](int slot) {
    unsigned inv = invmask[slot];
    if (inv) {
      AS3 char* zb = lds3 + slot * SLOT;
      #pragma unroll
      for (int u = 0; u < CPT; ++u)
        if (inv & (1u << u)) {
          typedef int i32x4 __attribute__((ext_vector_type(4)));
          *(AS3 i32x4*)(zb + ch_ldsoff[u]) = (i32x4){0, 0, 0, 0};
        }
    }
  };

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  long nkt = (M - m_begin + 63) / 64;
  if (nkt > steps_per_wg) nkt = steps_per_wg;
  if (nkt <= 0) nkt = 0;

  const int lbase =
      ((((lane >> 4) * 8 + ((lane & 15) >> 2)) * 16) +
       ((lane & 3) ^ (((lane >> 4) & 1) << 1)) * 4) * 2;

  // row-major [64 m][16 c] blocks: block stride 2 KB, half-kstep (32 m)
  // stride 32*16*2 = 1 KB
  auto trread8 = [&](AS3 char* region, int blk, int mhalf) -> bf16x8 {
    AS3 s16x4* p = (AS3 s16x4*)(region + blk * 2048 + mhalf * 1024 + lbase);
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
    if (pf < nkt) stage(pf, m_begin + (long)pf * 64);
  constexpr int INFLIGHT = (DEPTH - 1) * CPT;

  for (long kt = 0; kt < nkt; ++kt) {
    if (kt + DEPTH - 1 < nkt) {
      stage((int)((kt + DEPTH - 1) % DEPTH), m_begin + (kt + DEPTH - 1) * 64);
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(INFLIGHT) : "memory");
    } else {
      // pipeline tail: fewer than DEPTH-1 newer slots in flight, so a
      // counted wait would pass before this slot's loads retire (race)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    zero_invalid((int)(kt % DEPTH));
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __syncthreads();

    AS3 char* slot = lds3 + (kt % DEPTH) * SLOT;
    #pragma unroll
    for (int mh = 0; mh < 2; ++mh) {
      bf16x8 afrag[4], bfrag[4];
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        afrag[i] = trread8(slot, wr * 4 + i, mh);
        bfrag[i] = trread8(slot + REGB, wc * 4 + i, mh);
      }
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int i = 0; i < 4; ++i)
        #pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    // drain LDS reads before the barrier: the compiler may sink the mfma
    // consumers (and their lgkm waits) past __syncthreads, leaving tr-reads
    // in flight while the next iteration's staging overwrites the slot
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __syncthreads();
  }

  float* out = ws + (long)sp * Cout * Cin;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  #pragma unroll
  for (int j = 0; j < 4; ++j) {
    int cin_g = cin0 + wc * 64 + j * 16 + ccol;
    if (cin_g >= Cin) continue;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int cout_g0 = cout0 + wr * 64 + i * 16 + crow0;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        if (cout_g0 + r >= Cout) continue;
        out[(long)(cout_g0 + r) * Cin + cin_g] = acc[i][j][r];
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
  int ntn = (Cin + 63) / 64;
  int ntiles = ((Cout + 63) / 64) * ntn;
  long steps = (M + 31) / 32;
  if (taps == 1) {  // TN GEMM kernel: 128x128 tiles, 64-pixel steps
    ntn = (Cin + 127) / 128;
    ntiles = ((Cout + 127) / 128) * ntn;
    steps = (M + 63) / 64;
  }
  const int spw = (int)((steps + split - 1) / split);
  dim3 grid(ntiles * split), block(256);
  if (taps == 9)
    hipLaunchKernelGGL((conv_wrw2_kernel<9, 2>), grid, block, 0, s,
                       (const bf16_t*)dy, (const bf16_t*)x,
                       (const bf16_t*)guard, ws, N, H, W, Cin, Cout, OH, OW,
                       S_f, stride, P, Cin, spw, M, magicOW, magicOHOW, ntn,
                       split);
  else if (taps == 7)  // stem wrw over the NHWC4 padded view (pixstride 4)
    hipLaunchKernelGGL((conv_wrw2_kernel<7, 2>), grid, block, 0, s,
                       (const bf16_t*)dy, (const bf16_t*)x,
                       (const bf16_t*)guard, ws, N, H, W, Cin, Cout, OH, OW,
                       S_f, stride, P, 4, spw, M, magicOW, magicOHOW, ntn,
                       split);
  else if (taps == 1)
    hipLaunchKernelGGL((wrw_gemm_tn_kernel<2>), grid, block, 0, s,
                       (const bf16_t*)dy, (const bf16_t*)x,
                       (const bf16_t*)guard, ws, N, H, W, Cin, Cout, OH, OW,
                       stride, spw, M, magicOW, magicOHOW, ntn, split);
  const long n = (long)Cout * taps * Cin;
  dim3 rg((n + 255) / 256), rb(256);
  hipLaunchKernelGGL(wrw_reduce_kernel, rg, rb, 0, s, ws, dW, n, split);
}

}  // extern "C"
