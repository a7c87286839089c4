// MFMA bf16 GEMM for gfx950:  C[M,N] = A[M,K] @ B[N,K]^T  (both row-major,
// K-contiguous — the layout of torch Linear (x @ W^T) and of 1x1 convolutions
// on channels_last tensors, where A = the activation matrix [N*H*W, Cin] and
// B = the weight [Cout, Cin]).
//
// Structure (CDNA4 canonical, cf. the 128^2-tile anatomy of the CDNA HIP
// guide §5): 256 threads = 4 waves in a WGM x WGN grid, each wave owning a
// 64x64 output block as 4x4 fragments of v_mfma_f32_16x16x32_bf16
// accumulating in AGPRs. K steps of 32: operand tiles staged to LDS with
// __builtin_amdgcn_global_load_lds (16 B/lane, no VGPR round trip),
// double-buffered; fragment reads are ds_read_b128 (8 bf16/lane).
//
// Tile shapes: 128x128 (square GEMMs) and 256x64 (skinny-N: conv1x1 layers
// with Cout == 64 would waste half of a 128-wide tile).
//
// LDS addresses are XOR-swizzled (byte ^= ((byte>>7)&7)<<4, an involution
// preserving 16B chunks) so 16-lane column reads spread across banks (~2-way
// aliasing, free on CDNA4) instead of 8-way conflicting. global_load_lds
// writes linearly (wave-uniform base + lane*16), so the swizzle is applied by
// pre-permuting each lane's *global source* address and reading LDS through
// the same XOR — both-sides-or-neither.
//
// blockIdx -> tile mapping is XCD-aware (8 XCDs with private L2 on MI355X):
// contiguous grid chunks land on one XCD so neighboring tiles share L2.
#include "tfosr_common.h"
#include <type_traits>
#include <cstdlib>
#include <cstring>

typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define BK 32
#define AS_LDS(T) __attribute__((address_space(3))) T
#define LDS_SWZ(l) ((l) ^ ((((l) >> 7) & 7) << 4))

__device__ __forceinline__ void stage16(const char* src,
                                        __attribute__((address_space(3))) char* dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)src,
      (__attribute__((address_space(3))) void*)dst, 16, 0, 0);
}

template <typename OT, int WGM, int WGN, bool ACC = false>
__global__ __launch_bounds__(256, 2) void gemm_bt_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    OT* __restrict__ C, int M, int N, int K) {
  constexpr int BM = WGM * 64;
  constexpr int BN = WGN * 64;
  constexpr int ABYTES = BM * BK * 2;  // 64 B per row
  constexpr int BBYTES = BN * BK * 2;
  constexpr int ACHUNK = ABYTES / 16 / 256;  // 16B chunks per thread
  constexpr int BCHUNK = BBYTES / 16 / 256;
  __shared__ char lds[2 * (ABYTES + BBYTES)];
  __attribute__((address_space(3))) char* lds3 =
      (__attribute__((address_space(3))) char*)lds;

  const int ntn = (N + BN - 1) / BN;
  const int nwg = gridDim.x;
  // bijective XCD swizzle: chunk the grid so each XCD sees contiguous tiles
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
  const int wr = wave / WGN, wc = wave % WGN;

  // staging source addresses: dest is linear (chunk per thread), source gets
  // the (involutive) swizzle so swizzled-read sees the right bytes
  long srcA[ACHUNK], srcB[BCHUNK];
  const long Kb = (long)K * 2;
  #pragma unroll
  for (int u = 0; u < ACHUNK; ++u) {
    int d = (t + u * 256) * 16;
    int sl = LDS_SWZ(d);
    long row = sl >> 6;
    srcA[u] = ((tile_m + row < M) ? (tile_m + row) : (M - 1)) * Kb + (sl & 63);
  }
  #pragma unroll
  for (int u = 0; u < BCHUNK; ++u) {
    int d = (t + u * 256) * 16;
    int sl = LDS_SWZ(d);
    long row = sl >> 6;
    srcB[u] = ((tile_n + row < N) ? (tile_n + row) : (N - 1)) * Kb + (sl & 63);
  }

  auto stage_tile = [&](int buf, int kt) {
    const long kbyte = (long)kt * BK * 2;
    __attribute__((address_space(3))) char* la = lds3 + buf * (ABYTES + BBYTES);
    __attribute__((address_space(3))) char* lb = la + ABYTES;
    #pragma unroll
    for (int u = 0; u < ACHUNK; ++u)
      stage16((const char*)A + srcA[u] + kbyte, la + (t + u * 256) * 16);
    #pragma unroll
    for (int u = 0; u < BCHUNK; ++u)
      stage16((const char*)B + srcB[u] + kbyte, lb + (t + u * 256) * 16);
  };

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nkt = K / BK;
  int cur = 0;
  stage_tile(0, 0);

  const int frow = lane & 15;       // fragment row (m or n)
  const int kslot = lane >> 4;      // which 8-element K slice

  for (int kt = 0; kt < nkt; ++kt) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (kt + 1 < nkt) stage_tile(cur ^ 1, kt + 1);

    __attribute__((address_space(3))) char* la = lds3 + cur * (ABYTES + BBYTES);
    __attribute__((address_space(3))) char* lb = la + ABYTES;

    bf16x8 afrag[4], bfrag[4];
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      int l = (wr * 64 + mi * 16 + frow) * 64 + kslot * 16;
      afrag[mi] = *(__attribute__((address_space(3))) bf16x8*)(la + LDS_SWZ(l));
    }
    #pragma unroll
    for (int nj = 0; nj < 4; ++nj) {
      int l = (wc * 64 + nj * 16 + frow) * 64 + kslot * 16;
      bfrag[nj] = *(__attribute__((address_space(3))) bf16x8*)(lb + LDS_SWZ(l));
    }
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      #pragma unroll
      for (int nj = 0; nj < 4; ++nj)
        acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[mi], bfrag[nj], acc[mi][nj], 0, 0, 0);

    cur ^= 1;
    __syncthreads();
  }

  // epilogue: C/D layout col = lane&15, row = (lane>>4)*4 + reg
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  if constexpr (ACC && std::is_same<OT, bf16_t>::value) {
    // shuffle epilogue: the mfma lane layout reads/writes C in scattered
    // 32 B runs — fine for pure stores (write combining) but a read-modify-
    // write at that pattern costs as much as a separate eager add. Stage the
    // tile through LDS (8 KB bf16 per wave, reusing the staging ring) and do
    // the += with fully-coalesced 16 B-per-lane loads/stores.
    __syncthreads();  // staging LDS ring is dead now
    AS_LDS(char*) myl = lds3 + wave * 8192;
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        #pragma unroll
        for (int nj = 0; nj < 4; ++nj) {
          bf16_t v = (bf16_t)acc[mi][nj][r];
          short sv;
          __builtin_memcpy(&sv, &v, 2);
          *(AS_LDS(short*))(myl +
              ((mi * 16 + crow0 + r) * 64 + nj * 16 + ccol) * 2) = sv;
        }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    typedef short s16x8 __attribute__((ext_vector_type(8)));
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      int c = i * 64 + lane;          // 512 16B-chunks: row = c>>3, col8 = c&7
      int row = c >> 3, col0 = (c & 7) * 8;
      long m = tile_m + wr * 64 + row;
      long n = tile_n + wc * 64 + col0;
      if (m >= M || n + 7 >= N) {     // slow scalar tail at ragged edges
        if (m < M)
          for (int j = 0; j < 8 && n + j < N; ++j) {
            short sv = *(AS_LDS(short*))(myl + (row * 64 + col0 + j) * 2);
            bf16_t v;
            __builtin_memcpy(&v, &sv, 2);
            C[m * N + n + j] =
                (OT)((float)C[m * N + n + j] + (float)v);
          }
        continue;
      }
      s16x8 add = *(AS_LDS(s16x8*))(myl + (row * 64 + col0) * 2);
      s16x8 old = *(s16x8*)((char*)C + (m * N + n) * 2);
      s16x8 out;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        short sa = add[j], sb = old[j];
        bf16_t a, b;
        __builtin_memcpy(&a, &sa, 2);
        __builtin_memcpy(&b, &sb, 2);
        bf16_t o = (bf16_t)((float)a + (float)b);
        short so;
        __builtin_memcpy(&so, &o, 2);
        out[j] = so;
      }
      *(s16x8*)((char*)C + (m * N + n) * 2) = out;
    }
    return;
  }
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      long m = tile_m + wr * 64 + mi * 16 + crow0 + r;
      if (m >= M) continue;
      #pragma unroll
      for (int nj = 0; nj < 4; ++nj) {
        long n = tile_n + wc * 64 + nj * 16 + ccol;
        if (n < N)
          C[m * N + n] = ACC ? (OT)((float)C[m * N + n] + acc[mi][nj][r])
                             : (OT)acc[mi][nj][r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 256x256-tile, 8-wave, 4-deep-pipelined variant.
//
// The 2-buffer kernel above drains vmcnt to 0 at every K-step barrier — the
// classic CDNA4 stall (the s_waitcnt vmcnt(0) before s_barrier serializes the
// staging queue against compute). This variant keeps a 4-deep LDS ring of
// K-step tiles (4 x 32 KB = 128 KB) and waits with a *counted* vmcnt:
//
//   step t: s_waitcnt vmcnt(8); s_barrier;       // tile t's 4 loads landed
//           stage(t+3) -> ring[(t+3)%4];          // overwrites ring slot of
//                                                  // t-1, fully read last step
//           12 x ds_read_b128 frags; 32 MFMA (setprio-wrapped); s_barrier;
//
// Race-freedom: loads issued at step t target ring[(t+3)%4] == ring[(t-1)%4],
// whose reads completed at step t-1's closing barrier; vmcnt completion is
// in-order per thread, so "<= 8 outstanding" proves the 4 oldest (tile t's)
// have written LDS. No vmcnt(0) anywhere in the main loop.
// ---------------------------------------------------------------------------

#define B2_BM 256
#define B2_BN 256
#define B2_TILE_BYTES (B2_BM * BK * 2)   // 16 KB per operand per K-step
#define B2_SLOT (2 * B2_TILE_BYTES)      // A+B per ring slot

template <typename OT, bool ACC = false>
__global__ __launch_bounds__(512, 1) void gemm_bt256_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    OT* __restrict__ C, int M, int N, int K) {
  __shared__ char lds[4 * B2_SLOT];  // 128 KB ring
  __attribute__((address_space(3))) char* lds3 =
      (__attribute__((address_space(3))) char*)lds;

  const int ntn = (N + B2_BN - 1) / B2_BN;
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const long tile_m = (long)(wgid / ntn) * B2_BM;
  const long tile_n = (long)(wgid % ntn) * B2_BN;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;          // 8 waves: 2 (M) x 4 (N)
  const int wr = wave >> 2;         // 0..1 -> 128 rows each
  const int wc = wave & 3;          // 0..3 -> 64 cols each

  // staging: 4 x 16B chunks per thread per K-step (2 for A, 2 for B)
  const long Kb = (long)K * 2;
  long srcA[2], srcB[2];
  #pragma unroll
  for (int u = 0; u < 2; ++u) {
    int d = (t + u * 512) * 16;
    int sl = LDS_SWZ(d);
    long row = sl >> 6;
    srcA[u] = ((tile_m + row < M) ? (tile_m + row) : (M - 1)) * Kb + (sl & 63);
    srcB[u] = ((tile_n + row < N) ? (tile_n + row) : (N - 1)) * Kb + (sl & 63);
  }

  auto stage_tile = [&](int slot, int kt) {
    const long kbyte = (long)kt * BK * 2;
    __attribute__((address_space(3))) char* la = lds3 + slot * B2_SLOT;
    __attribute__((address_space(3))) char* lb = la + B2_TILE_BYTES;
    #pragma unroll
    for (int u = 0; u < 2; ++u)
      stage16((const char*)A + srcA[u] + kbyte, la + (t + u * 512) * 16);
    #pragma unroll
    for (int u = 0; u < 2; ++u)
      stage16((const char*)B + srcB[u] + kbyte, lb + (t + u * 512) * 16);
  };

  f32x4 acc[8][4];
  #pragma unroll
  for (int i = 0; i < 8; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nkt = K / BK;
  stage_tile(0, 0);
  if (1 < nkt) stage_tile(1, 1);
  if (2 < nkt) stage_tile(2, 2);

  const int frow = lane & 15;
  const int kslot = lane >> 4;

  for (int kt = 0; kt < nkt; ++kt) {
    if (kt + 2 < nkt)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else  // tail: fewer newer tiles in flight than the counted wait assumes
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __syncthreads();
    if (kt + 3 < nkt) stage_tile((kt + 3) & 3, kt + 3);

    __attribute__((address_space(3))) char* la = lds3 + (kt & 3) * B2_SLOT;
    __attribute__((address_space(3))) char* lb = la + B2_TILE_BYTES;

    bf16x8 afrag[8], bfrag[4];
    #pragma unroll
    for (int nj = 0; nj < 4; ++nj) {
      int l = (wc * 64 + nj * 16 + frow) * 64 + kslot * 16;
      bfrag[nj] = *(__attribute__((address_space(3))) bf16x8*)(lb + LDS_SWZ(l));
    }
    #pragma unroll
    for (int mi = 0; mi < 8; ++mi) {
      int l = (wr * 128 + mi * 16 + frow) * 64 + kslot * 16;
      afrag[mi] = *(__attribute__((address_space(3))) bf16x8*)(la + LDS_SWZ(l));
    }
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int mi = 0; mi < 8; ++mi)
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
  for (int mi = 0; mi < 8; ++mi) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      long m = tile_m + wr * 128 + mi * 16 + crow0 + r;
      if (m >= M) continue;
      #pragma unroll
      for (int nj = 0; nj < 4; ++nj) {
        long n = tile_n + wc * 64 + nj * 16 + ccol;
        if (n < N)
          C[m * N + n] = ACC ? (OT)((float)C[m * N + n] + acc[mi][nj][r])
                             : (OT)acc[mi][nj][r];
      }
    }
  }
}

// Raw-intrinsic probe: feed per-lane fragments, observe the accumulator —
// lets the host test discover/verify the A/B/C lane->element mappings.
__global__ void mfma_probe_kernel(const short* __restrict__ a,
                                  const short* __restrict__ b,
                                  float* __restrict__ c) {
  int l = threadIdx.x;
  bf16x8 av = *(const bf16x8*)(a + l * 8);
  bf16x8 bv = *(const bf16x8*)(b + l * 8);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 4; ++r) c[l * 4 + r] = acc[r];
}

extern "C" {

void tfosr_gemm_bt_acc(const void*, const void*, void*, int, int, int, int,
                       int, hipStream_t);

void tfosr_gemm_bt(const void* A, const void* B, void* C, int out_bf16,
                   int M, int N, int K, hipStream_t s) {
  tfosr_gemm_bt_acc(A, B, C, out_bf16, M, N, K, 0, s);
}

// accum != 0: C += A @ B^T (the ResNet residual-join gradient accumulation
// fused into the conv1x1 dgrad epilogue instead of a separate eager add)
void tfosr_gemm_bt_acc(const void* A, const void* B, void* C, int out_bf16,
                       int M, int N, int K, int accum, hipStream_t s) {
  // 256^2 4-deep-pipelined kernel for large-K tiles (917 TF @4096^3 vs 644
  // for the 2-buffer kernel; the deep prologue loses on the skinny-K conv
  // shapes, which stay on the 128^2/256x64 kernels). Toggle: TFOS_GEMM256=off
  static int use256 = -1;
  if (use256 < 0) {
    const char* e = getenv("TFOS_GEMM256");
    use256 = (e == nullptr || strcmp(e, "off") != 0) ? 1 : 0;
  }
  if (use256 && M >= 1024 && N >= 192 && K >= 1024) {
    int ntm = (M + B2_BM - 1) / B2_BM, ntn = (N + B2_BN - 1) / B2_BN;
    dim3 grid(ntm * ntn);
    if (out_bf16) {
      if (accum)
        hipLaunchKernelGGL((gemm_bt256_kernel<bf16_t, true>), grid, dim3(512),
                           0, s, (const bf16_t*)A, (const bf16_t*)B,
                           (bf16_t*)C, M, N, K);
      else
        hipLaunchKernelGGL((gemm_bt256_kernel<bf16_t, false>), grid, dim3(512),
                           0, s, (const bf16_t*)A, (const bf16_t*)B,
                           (bf16_t*)C, M, N, K);
    } else {
      hipLaunchKernelGGL((gemm_bt256_kernel<float, false>), grid, dim3(512), 0,
                         s, (const bf16_t*)A, (const bf16_t*)B, (float*)C, M,
                         N, K);
    }
    return;
  }
  // skinny-N tile when it reduces waste (Cout=64 conv1x1 layers)
  const bool skinny = (N % 128 != 0) && (N % 64 == 0 || N <= 64);
  if (skinny) {
    int ntm = (M + 255) / 256, ntn = (N + 63) / 64;
    dim3 grid(ntm * ntn);
    if (out_bf16) {
      if (accum)
        hipLaunchKernelGGL((gemm_bt_kernel<bf16_t, 4, 1, true>), grid,
                           dim3(256), 0, s, (const bf16_t*)A,
                           (const bf16_t*)B, (bf16_t*)C, M, N, K);
      else
        hipLaunchKernelGGL((gemm_bt_kernel<bf16_t, 4, 1, false>), grid,
                           dim3(256), 0, s, (const bf16_t*)A,
                           (const bf16_t*)B, (bf16_t*)C, M, N, K);
    } else {
      hipLaunchKernelGGL((gemm_bt_kernel<float, 4, 1>), grid, dim3(256), 0, s,
                         (const bf16_t*)A, (const bf16_t*)B, (float*)C, M, N,
                         K);
    }
    return;
  }
  int ntm = (M + 127) / 128, ntn = (N + 127) / 128;
  dim3 grid(ntm * ntn);
  if (out_bf16) {
    if (accum)
      hipLaunchKernelGGL((gemm_bt_kernel<bf16_t, 2, 2, true>), grid, dim3(256),
                         0, s, (const bf16_t*)A, (const bf16_t*)B, (bf16_t*)C,
                         M, N, K);
    else
      hipLaunchKernelGGL((gemm_bt_kernel<bf16_t, 2, 2, false>), grid,
                         dim3(256), 0, s, (const bf16_t*)A, (const bf16_t*)B,
                         (bf16_t*)C, M, N, K);
  } else {
    hipLaunchKernelGGL((gemm_bt_kernel<float, 2, 2>), grid, dim3(256), 0, s,
                       (const bf16_t*)A, (const bf16_t*)B, (float*)C, M, N, K);
  }
}

void tfosr_mfma_probe(const short* a, const short* b, float* c, hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, s, a, b, c);
}

}  // extern "C"
