// NHWC (channels_last) MaxPool2d fwd/bwd for gfx950 — bf16/f32, vectorized
// over channels. The ResNet stem's 3x3/s2 pool via torch's nhwc kernel cost
// ~30% of a BN pass's roofline; this one is a straight HBM-bound streaming op:
//   fwd: each thread owns 8 (bf16) channels of one output pixel; reads the
//        KxK window rows as 16 B vectors, writes max + a packed window-index.
//   bwd: gather formulation (no atomics): each *input* pixel checks the <=
//        ceil(K/S)^2 windows that could have selected it and sums their dy.
#include "tfosr_common.h"

typedef unsigned char u8;
typedef unsigned int u32;

template <typename T, int V>
__device__ __forceinline__ void VecIO_load(const T* p, float* out);

template <>
__device__ __forceinline__ void VecIO_load<bf16_t, 8>(const bf16_t* p, float* out) {
  s8v v = *(const s8v*)p;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    union { short s; bf16_t b; } u;
    u.s = v[j];
    out[j] = (float)u.b;
  }
}

template <>
__device__ __forceinline__ void VecIO_load<float, 4>(const float* p, float* out) {
  f4v v = *(const f4v*)p;
  #pragma unroll
  for (int j = 0; j < 4; ++j) out[j] = v[j];
}


template <typename T, int V>
__global__ void maxpool_fwd_nhwc(const T* __restrict__ x, T* __restrict__ y,
                                 u8* __restrict__ idx, int N, int C, int H,
                                 int W, int OH, int OW, int K, int S, int P) {
  const u32 Cv = C / V;
  const u32 total = (u32)N * OH * OW * Cv;
  for (u32 i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (u32)gridDim.x * blockDim.x) {
    u32 cv = i % Cv;
    u32 rest = i / Cv;
    u32 ow = rest % OW;
    rest /= OW;
    u32 oh = rest % OH;
    u32 n = rest / OH;
    float best[V];
    int bidx[V];
    #pragma unroll
    for (int j = 0; j < V; ++j) { best[j] = -INFINITY; bidx[j] = 0; }
    int ih0 = (int)oh * S - P, iw0 = (int)ow * S - P;
    for (int ky = 0; ky < K; ++ky) {
      int ih = ih0 + ky;
      if (ih < 0 || ih >= H) continue;
      for (int kx = 0; kx < K; ++kx) {
        int iw = iw0 + kx;
        if (iw < 0 || iw >= W) continue;
        const T* p = x + (((size_t)n * H + ih) * W + iw) * C + cv * V;
        float v[V];
        VecIO_load<T, V>(p, v);
        #pragma unroll
        for (int j = 0; j < V; ++j) {
          if (v[j] > best[j]) { best[j] = v[j]; bidx[j] = ky * K + kx; }
        }
      }
    }
    T* py = y + (((size_t)n * OH + oh) * OW + ow) * C + cv * V;
    u8* pi = idx + (((size_t)n * OH + oh) * OW + ow) * C + cv * V;
    #pragma unroll
    for (int j = 0; j < V; ++j) {
      py[j] = (T)best[j];
      pi[j] = (u8)bidx[j];
    }
  }
}

template <typename T, int V>
__global__ void maxpool_bwd_nhwc(const T* __restrict__ dy,
                                 const u8* __restrict__ idx, T* __restrict__ dx,
                                 int N, int C, int H, int W, int OH, int OW,
                                 int K, int S, int P) {
  const u32 Cv = C / V;
  const u32 total = (u32)N * H * W * Cv;
  for (u32 i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (u32)gridDim.x * blockDim.x) {
    u32 cv = i % Cv;
    u32 rest = i / Cv;
    u32 iw = rest % W;
    rest /= W;
    u32 ih = rest % H;
    u32 n = rest / H;
    float acc[V];
    #pragma unroll
    for (int j = 0; j < V; ++j) acc[j] = 0.f;
    // windows (oh, ow) with oh*S - P <= ih < oh*S - P + K
    int oh_lo = ((int)ih + P - K + S) / S;  // ceil((ih+P-K+1)/S)
    if (oh_lo < 0) oh_lo = 0;
    int oh_hi = ((int)ih + P) / S;
    if (oh_hi >= OH) oh_hi = OH - 1;
    int ow_lo = ((int)iw + P - K + S) / S;
    if (ow_lo < 0) ow_lo = 0;
    int ow_hi = ((int)iw + P) / S;
    if (ow_hi >= OW) ow_hi = OW - 1;
    for (int oh = oh_lo; oh <= oh_hi; ++oh) {
      int ky = (int)ih - (oh * S - P);
      if (ky < 0 || ky >= K) continue;
      for (int ow = ow_lo; ow <= ow_hi; ++ow) {
        int kx = (int)iw - (ow * S - P);
        if (kx < 0 || kx >= K) continue;
        const size_t off = (((size_t)n * OH + oh) * OW + ow) * C + cv * V;
        const u8 want = (u8)(ky * K + kx);
        float v[V];
        VecIO_load<T, V>(dy + off, v);
        #pragma unroll
        for (int j = 0; j < V; ++j)
          if (idx[off + j] == want) acc[j] += v[j];
      }
    }
    T* p = dx + (((size_t)n * H + ih) * W + iw) * C + cv * V;
    #pragma unroll
    for (int j = 0; j < V; ++j) p[j] = (T)acc[j];
  }
}

extern "C" {

void tfosr_maxpool_fwd(const void* x, void* y, unsigned char* idx, int is_bf16,
                       int N, int C, int H, int W, int OH, int OW, int K,
                       int S, int P, hipStream_t s) {
  long total = (long)N * OH * OW * C / (is_bf16 ? 8 : 4);
  int grid = tfosr_grid(total, 256);
  if (is_bf16)
    hipLaunchKernelGGL((maxpool_fwd_nhwc<bf16_t, 8>), dim3(grid), dim3(256), 0, s,
                       (const bf16_t*)x, (bf16_t*)y, idx, N, C, H, W, OH, OW,
                       K, S, P);
  else
    hipLaunchKernelGGL((maxpool_fwd_nhwc<float, 4>), dim3(grid), dim3(256), 0, s,
                       (const float*)x, (float*)y, idx, N, C, H, W, OH, OW,
                       K, S, P);
}

void tfosr_maxpool_bwd(const void* dy, const unsigned char* idx, void* dx,
                       int is_bf16, int N, int C, int H, int W, int OH, int OW,
                       int K, int S, int P, hipStream_t s) {
  long total = (long)N * H * W * C / (is_bf16 ? 8 : 4);
  int grid = tfosr_grid(total, 256);
  if (is_bf16)
    hipLaunchKernelGGL((maxpool_bwd_nhwc<bf16_t, 8>), dim3(grid), dim3(256), 0, s,
                       (const bf16_t*)dy, idx, (bf16_t*)dx, N, C, H, W, OH, OW,
                       K, S, P);
  else
    hipLaunchKernelGGL((maxpool_bwd_nhwc<float, 4>), dim3(grid), dim3(256), 0, s,
                       (const float*)dy, idx, (float*)dx, N, C, H, W, OH, OW,
                       K, S, P);
}

}  // extern "C"
