// Common device helpers for the tfosr CDNA4 (gfx950) kernels.
// Wave size on CDNA is 64 (not 32); hard-code per the CDNA4 programming model.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define TFOSR_WAVE 64
#define TFOSR_CU 256          // MI355X compute units
#define TFOSR_MAX_GRID 2048   // memory-bound grid cap: ~8 blocks/CU, grid-stride rest

typedef __hip_bfloat16 bf16_t;

// short8 = 8 bf16 (16 B) — the coalescing sweet spot for bf16 loads.
typedef short s8v __attribute__((ext_vector_type(8)));
typedef float f4v __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf2f(bf16_t v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16_t f2bf(float v) { return __float2bfloat16(v); }

// wave-wide sum over all 64 lanes
__device__ __forceinline__ float wave_sum(float v) {
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}
__device__ __forceinline__ float wave_max(float v) {
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// block-level sum reduction (block size multiple of 64, <= 1024)
template <int BLOCK>
__device__ __forceinline__ float block_sum(float v, float* scratch) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  v = wave_sum(v);
  if (lane == 0) scratch[wave] = v;
  __syncthreads();
  constexpr int NW = BLOCK / 64;
  v = (threadIdx.x < NW) ? scratch[threadIdx.x] : 0.f;
  if (wave == 0) {
    for (int off = NW / 2; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  }
  return v;  // valid in thread 0
}

template <int BLOCK>
__device__ __forceinline__ float block_max(float v, float* scratch) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  v = wave_max(v);
  if (lane == 0) scratch[wave] = v;
  __syncthreads();
  constexpr int NW = BLOCK / 64;
  v = (threadIdx.x < NW) ? scratch[threadIdx.x] : -INFINITY;
  if (wave == 0) {
    for (int off = NW / 2; off > 0; off >>= 1)
      v = fmaxf(v, __shfl_down(v, off, 64));
  }
  return v;
}

static inline int tfosr_grid(long total, int block, int cap = TFOSR_MAX_GRID) {
  long g = (total + block - 1) / block;
  return (int)(g < cap ? (g > 0 ? g : 1) : cap);
}
