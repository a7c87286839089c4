// Convolution weight-gradient (wrw) as an MFMA TN contraction on gfx950:
//
//   dW[cout, (r,s), cin] = sum_m dy[m, cout] * x_shift(r,s)[m, cin]
//
// Both operands are M-major in memory (rows = output pixels), but MFMA wants
// per-lane K(=m)-contiguous fragments — so tiles are staged to LDS
// *transposed* via registers (global short8 loads, scattered ds_write_b16),
// XOR-swizzled against the 128 B-row bank conflict, then consumed with clean
// ds_read_b128. The m-loop is double-buffered issue-early/write-late: the
// next block's global loads are issued before this block's MFMA, and the LDS
// write happens after the barrier (HBM latency hides under compute).
//
// The (r,s) taps of a 3x3 are independent slices of dW: gridDim encodes
// (output tile, tap, M-split); partial products accumulate into fp32 dW with
// one atomicAdd per element per split. Padding taps stage zeros (reg-staged,
// so no guard buffer is needed). 1x1 wrw is the same kernel with R=S=1, P=0.
#include "tfosr_common.h"

typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define WW_BM 128        // Cout tile
#define WW_BN 128        // Cin tile
#define WW_BK 64         // m rows staged per block-iter
#define WW_SWZ(l) ((l) ^ ((((l) >> 7) & 7) << 4))
#define WW_TILE (WW_BM * WW_BK * 2)  // 16 KB per transposed operand tile

// transposed LDS write of one short8: value[j] -> lds[(c0+j)][m]
__device__ __forceinline__ void ww_write8(
    __attribute__((address_space(3))) char* base, int c0, int m, bf16x8 v) {
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    int l = ((c0 + j) * WW_BK + m) * 2;
    *(__attribute__((address_space(3))) short*)(base + WW_SWZ(l)) = v[j];
  }
}

__global__ __launch_bounds__(256, 2) void conv_wrw_kernel(
    const bf16_t* __restrict__ dy,  // [M, Cout] (channels_last view)
    const bf16_t* __restrict__ X,   // [N, H, W, Cin] (channels_last storage)
    float* __restrict__ dW,         // [Cout, R*S*Cin], pre-zeroed
    int Nn, int H, int Wd, int Cin, int Cout, int OH, int OW,
    int R, int S, int P, int Cstride, int nsplit) {
  __shared__ char lds[4 * WW_TILE];  // [buf][dyT | xT]
  __attribute__((address_space(3))) char* lds3 =
      (__attribute__((address_space(3))) char*)lds;

  const long M = (long)Nn * OH * OW;
  const int ntm = (Cout + WW_BM - 1) / WW_BM;
  const int ntn = (Cin + WW_BN - 1) / WW_BN;
  const int ntiles = ntm * ntn;
  // blockIdx = (((split * RS) + tap) * ntiles) + tile
  int id = blockIdx.x;
  const int tile = id % ntiles;
  id /= ntiles;
  const int tap = id % (R * S);
  const int split = id / (R * S);
  const int rr = tap / S, ss = tap % S;
  const long tile_co = (long)(tile / ntn) * WW_BM;
  const long tile_ci = (long)(tile % ntn) * WW_BN;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;  // 2x2 waves, 64x64 each

  // m-range for this split
  const long mblocks = (M + WW_BK - 1) / WW_BK;
  const long per = (mblocks + nsplit - 1) / nsplit;
  const long mb0 = split * per;
  const long mb1 = (mb0 + per < mblocks) ? (mb0 + per) : mblocks;
  if (mb0 >= mb1) return;

  // this thread's staged chunks: 64 m-rows x 16 col-chunks (8 cols each)
  // = 1024 chunks per operand tile -> 4 per thread
  int cm[4], cc[4];
  #pragma unroll
  for (int u = 0; u < 4; ++u) {
    int chunk = t + u * 256;
    cm[u] = chunk >> 4;            // m row within block (0..63)
    cc[u] = (chunk & 15) * 8;      // col0 (0..120)
  }

  auto load_regs = [&](long mb, bf16x8* rdy, bf16x8* rx) {
    #pragma unroll
    for (int u = 0; u < 4; ++u) {
      long m = mb * WW_BK + cm[u];
      bf16x8 zero = {0, 0, 0, 0, 0, 0, 0, 0};
      rdy[u] = zero;
      rx[u] = zero;
      if (m < M) {
        long co = tile_co + cc[u];
        if (co < Cout)
          rdy[u] = *(const bf16x8*)(dy + m * Cout + co);
        long n = m / ((long)OH * OW);
        long rem = m - n * (long)OH * OW;
        int oh = (int)(rem / OW);
        int ow = (int)(rem - (long)oh * OW);
        int ih = oh + rr - P, iw = ow + ss - P;  // stride-1 wrw
        long ci = tile_ci + cc[u];
        if (ih >= 0 && ih < H && iw >= 0 && iw < Wd && ci < Cin)
          rx[u] = *(const bf16x8*)(X + (((long)n * H + ih) * Wd + iw) * Cin + ci);
      }
    }
  };

  auto write_lds = [&](int buf, const bf16x8* rdy, const bf16x8* rx) {
    __attribute__((address_space(3))) char* ld = lds3 + buf * 2 * WW_TILE;
    __attribute__((address_space(3))) char* lx = ld + WW_TILE;
    #pragma unroll
    for (int u = 0; u < 4; ++u) {
      ww_write8(ld, cc[u], cm[u], rdy[u]);
      ww_write8(lx, cc[u], cm[u], rx[u]);
    }
  };

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  bf16x8 rdyA[4], rxA[4], rdyB[4], rxB[4];
  load_regs(mb0, rdyA, rxA);
  write_lds(0, rdyA, rxA);

  const int frow = lane & 15;
  const int kslot = lane >> 4;

  for (long mb = mb0; mb < mb1; ++mb) {
    const bool more = mb + 1 < mb1;
    if (more) {
      // issue next block's global loads before compute (issue-early)
      if ((mb - mb0) & 1) load_regs(mb + 1, rdyA, rxA);
      else load_regs(mb + 1, rdyB, rxB);
    }
    __syncthreads();
    const int buf = (int)((mb - mb0) & 1);
    __attribute__((address_space(3))) char* ld = lds3 + buf * 2 * WW_TILE;
    __attribute__((address_space(3))) char* lx = ld + WW_TILE;
    #pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // two 32-m K-steps per 64-m block
      bf16x8 afrag[4], bfrag[4];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int l = ((wr * 64 + mi * 16 + frow) * WW_BK + ks * 32 + kslot * 8) * 2;
        afrag[mi] = *(__attribute__((address_space(3))) bf16x8*)(ld + WW_SWZ(l));
      }
      #pragma unroll
      for (int nj = 0; nj < 4; ++nj) {
        int l = ((wc * 64 + nj * 16 + frow) * WW_BK + ks * 32 + kslot * 8) * 2;
        bfrag[nj] = *(__attribute__((address_space(3))) bf16x8*)(lx + WW_SWZ(l));
      }
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int nj = 0; nj < 4; ++nj)
          acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[nj], acc[mi][nj], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
    if (more) {
      // write-late: the next block's regs land in the buffer just consumed
      if ((mb - mb0) & 1) write_lds(buf ^ 1, rdyA, rxA);
      else write_lds(buf ^ 1, rdyB, rxB);
    }
  }

  // accumulate into dW[cout][tap*Cin + cin] (fp32, one atomic per elem/split)
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      long co = tile_co + wr * 64 + mi * 16 + crow0 + r;
      if (co >= Cout) continue;
      #pragma unroll
      for (int nj = 0; nj < 4; ++nj) {
        long ci = tile_ci + wc * 64 + nj * 16 + ccol;
        if (ci < Cin)
          atomicAdd(&dW[co * (long)Cstride + tap * (long)Cin + ci],
                    acc[mi][nj][r]);
      }
    }
  }
}

extern "C" {

void tfosr_conv_wrw(const void* dy, const void* x, float* dW, int N, int H,
                    int W, int Cin, int Cout, int OH, int OW, int R, int S,
                    int P, hipStream_t s) {
  const long M = (long)N * OH * OW;
  const int ntm = (Cout + WW_BM - 1) / WW_BM;
  const int ntn = (Cin + WW_BN - 1) / WW_BN;
  const int ntiles = ntm * ntn;
  const int rs = R * S;
  long mblocks = (M + WW_BK - 1) / WW_BK;
  // enough splits to fill the chip, bounded by available m-blocks
  long want = (2048 + (long)ntiles * rs - 1) / ((long)ntiles * rs);
  int nsplit = (int)(want < 1 ? 1 : (want > mblocks ? mblocks : want));
  dim3 grid((unsigned)(ntiles * rs * nsplit));
  hipLaunchKernelGGL(conv_wrw_kernel, grid, dim3(256), 0, s,
                     (const bf16_t*)dy, (const bf16_t*)x, dW, N, H, W, Cin,
                     Cout, OH, OW, R, S, P, rs * Cin, nsplit);
}

}  // extern "C"
