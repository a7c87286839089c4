// MFMA bf16 GEMM for gfx950:  C[M,N] = A[M,K] @ B[N,K]^T  (both row-major,
// K-contiguous — the natural layout for torch.nn.Linear's x @ W^T).
//
// Structure (CDNA4 canonical, cf. the 128^2-tile anatomy in the CDNA HIP
// guide §5): 256 threads = 4 waves in a 2x2 grid, each wave owns a 64x64
// output block as 4x4 fragments of v_mfma_f32_16x16x32_bf16 accumulating in
// AGPRs. K steps of 32: A/B tiles staged to LDS with
// __builtin_amdgcn_global_load_lds (16 B/lane, no VGPR round trip),
// double-buffered; fragment reads are ds_read_b128 (8 bf16/lane).
//
// LDS addresses are XOR-swizzled (byte ^= ((byte>>7)&7)<<4, an involution that
// preserves 16B chunks) so the 16-lane column reads spread across banks
// (~2-way aliasing, which is free on CDNA4) instead of 8-way conflicting.
// Since global_load_lds writes linearly (wave-uniform base + lane*16), the
// swizzle is applied by pre-permuting each lane's *global source* address and
// reading LDS through the same XOR — both-sides-or-neither.
//
// blockIdx -> tile mapping is XCD-aware (8 XCDs with private L2 on MI355X):
// contiguous grid chunks land on one XCD so neighboring tiles share L2.
#include "tfosr_common.h"

typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define BM 128
#define BN 128
#define BK 32
#define TILE_BYTES (BM * BK * 2)  // 8192 B per operand tile
#define LDS_SWZ(l) ((l) ^ ((((l) >> 7) & 7) << 4))

__device__ __forceinline__ void stage16(const char* src,
                                        __attribute__((address_space(3))) char* dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)src,
      (__attribute__((address_space(3))) void*)dst, 16, 0, 0);
}

template <typename OT>
__global__ __launch_bounds__(256, 2) void gemm_bt_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    OT* __restrict__ C, int M, int N, int K) {
  __shared__ char lds[2 * 2 * TILE_BYTES];  // [buf][A|B][tile]
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
  const int wr = wave >> 1, wc = wave & 1;  // 2x2 wave grid

  // staging: thread t covers 16B chunks t and t+256 of each 8192B tile
  // dest is linear; source address gets the inverse (== same) swizzle
  long srcA_row[2], srcA_col[2], srcB_row[2], srcB_col[2];
  #pragma unroll
  for (int u = 0; u < 2; ++u) {
    int d = (t + u * 256) * 16;      // linear dest byte offset in tile
    int sl = LDS_SWZ(d);             // source linear position
    long row = sl >> 6;              // 64 B per row (32 bf16)
    long col = sl & 63;              // byte within row
    srcA_row[u] = (tile_m + row < M) ? (tile_m + row) : (M - 1);
    srcA_col[u] = col;
    srcB_row[u] = (tile_n + row < N) ? (tile_n + row) : (N - 1);
    srcB_col[u] = col;
  }

  const long Kb = (long)K * 2;  // row stride in bytes

  auto stage_tile = [&](int buf, int kt) {
    const long kbyte = (long)kt * BK * 2;
    __attribute__((address_space(3))) char* la = lds3 + buf * 2 * TILE_BYTES;
    __attribute__((address_space(3))) char* lb = la + TILE_BYTES;
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      int d = (t + u * 256) * 16;
      stage16((const char*)A + srcA_row[u] * Kb + kbyte + srcA_col[u], la + d);
      stage16((const char*)B + srcB_row[u] * Kb + kbyte + srcB_col[u], lb + d);
    }
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

    __attribute__((address_space(3))) char* la = lds3 + cur * 2 * TILE_BYTES;
    __attribute__((address_space(3))) char* lb = la + TILE_BYTES;

    bf16x8 afrag[4], bfrag[4];
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      int row = wr * 64 + mi * 16 + frow;
      int l = row * 64 + kslot * 16;
      afrag[mi] = *(__attribute__((address_space(3))) bf16x8*)(la + LDS_SWZ(l));
    }
    #pragma unroll
    for (int nj = 0; nj < 4; ++nj) {
      int row = wc * 64 + nj * 16 + frow;
      int l = row * 64 + kslot * 16;
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
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      long m = tile_m + wr * 64 + mi * 16 + crow0 + r;
      if (m >= M) continue;
      #pragma unroll
      for (int nj = 0; nj < 4; ++nj) {
        long n = tile_n + wc * 64 + nj * 16 + ccol;
        if (n < N) C[m * N + n] = (OT)acc[mi][nj][r];
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

void tfosr_gemm_bt(const void* A, const void* B, void* C, int out_bf16,
                   int M, int N, int K, hipStream_t s) {
  int ntm = (M + BM - 1) / BM, ntn = (N + BN - 1) / BN;
  dim3 grid(ntm * ntn);
  if (out_bf16)
    hipLaunchKernelGGL(gemm_bt_kernel<bf16_t>, grid, dim3(256), 0, s,
                       (const bf16_t*)A, (const bf16_t*)B, (bf16_t*)C, M, N, K);
  else
    hipLaunchKernelGGL(gemm_bt_kernel<float>, grid, dim3(256), 0, s,
                       (const bf16_t*)A, (const bf16_t*)B, (float*)C, M, N, K);
}

void tfosr_mfma_probe(const short* a, const short* b, float* c, hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, s, a, b, c);
}

}  // extern "C"
