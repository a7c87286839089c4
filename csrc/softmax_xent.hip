// Fused sparse softmax-cross-entropy for gfx950.
// fwd: one block per row -> loss[n] (fp32) and logsumexp[n] (fp32, max folded).
// bwd: dlogits = (softmax - onehot) * upstream, recomputed from (logits, lse).
#include "tfosr_common.h"

template <typename T>
__global__ void xent_fwd_kernel(const T* __restrict__ logits,
                                const long* __restrict__ target,
                                float* __restrict__ loss, float* __restrict__ lse,
                                int N, int C) {
  __shared__ float scratch[8];
  for (int n = blockIdx.x; n < N; n += gridDim.x) {
    const T* row = logits + (long)n * C;
    float m = -INFINITY;
    for (int c = threadIdx.x; c < C; c += blockDim.x)
      m = fmaxf(m, (float)row[c]);
    m = block_max<256>(m, scratch);
    if (threadIdx.x == 0) scratch[0] = m;
    __syncthreads();
    m = scratch[0];
    __syncthreads();
    float s = 0.f;
    for (int c = threadIdx.x; c < C; c += blockDim.x)
      s += __expf((float)row[c] - m);
    s = block_sum<256>(s, scratch);
    if (threadIdx.x == 0) {
      float l = m + __logf(s);
      lse[n] = l;
      loss[n] = l - (float)row[target[n]];
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void xent_bwd_kernel(const T* __restrict__ logits,
                                const float* __restrict__ lse,
                                const long* __restrict__ target,
                                const float* __restrict__ gout,
                                T* __restrict__ dlogits, long total, int C) {
  for (long idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    long n = idx / C;
    int c = (int)(idx - n * C);
    float p = __expf((float)logits[idx] - lse[n]);
    float d = (p - (target[n] == c ? 1.f : 0.f)) * gout[n];
    dlogits[idx] = (T)d;
  }
}

extern "C" {

void tfosr_xent_fwd(const void* logits, const long* target, float* loss,
                    float* lse, int is_bf16, int N, int C, hipStream_t s) {
  int grid = N < TFOSR_MAX_GRID ? (N > 0 ? N : 1) : TFOSR_MAX_GRID;
  if (is_bf16)
    hipLaunchKernelGGL(xent_fwd_kernel<bf16_t>, dim3(grid), dim3(256), 0, s,
                       (const bf16_t*)logits, target, loss, lse, N, C);
  else
    hipLaunchKernelGGL(xent_fwd_kernel<float>, dim3(grid), dim3(256), 0, s,
                       (const float*)logits, target, loss, lse, N, C);
}

void tfosr_xent_bwd(const void* logits, const float* lse, const long* target,
                    const float* gout, void* dlogits, int is_bf16, int N, int C,
                    hipStream_t s) {
  long total = (long)N * C;
  int grid = tfosr_grid(total, 256);
  if (is_bf16)
    hipLaunchKernelGGL(xent_bwd_kernel<bf16_t>, dim3(grid), dim3(256), 0, s,
                       (const bf16_t*)logits, lse, target, gout,
                       (bf16_t*)dlogits, total, C);
  else
    hipLaunchKernelGGL(xent_bwd_kernel<float>, dim3(grid), dim3(256), 0, s,
                       (const float*)logits, lse, target, gout,
                       (float*)dlogits, total, C);
}

}  // extern "C"
