// Elementwise / optimizer kernels for gfx950:
//  - nhwc_pack: uint8 NHWC batch -> normalized NCHW (or channels_last) f32/bf16
//  - sgd_step: fused SGD(momentum, weight_decay) over a flat parameter bucket
#include "tfosr_common.h"

// ---------------------------------------------------------------------------
// NHWC uint8 -> normalized tensor (the DataFeed->GPU ingest kernel)
// ---------------------------------------------------------------------------

typedef uint8_t u8;
typedef u8 u8x16 __attribute__((ext_vector_type(16)));

// channels_last output: pure elementwise in source order (fast path).
template <typename T>
__global__ void nhwc_pack_cl_kernel(const u8* __restrict__ in, T* __restrict__ out,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ std_,
                                    float scale, long total, int C) {
  constexpr int V = 16;
  const long nvec = total / V;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    const long base = i * V;
    u8x16 v = *(const u8x16*)(in + base);
    T o[V];
    #pragma unroll
    for (int j = 0; j < V; ++j) {
      int c = (int)((base + j) % C);
      o[j] = (T)(((float)v[j] * scale - mean[c]) / std_[c]);
    }
    #pragma unroll
    for (int j = 0; j < V; ++j) out[base + j] = o[j];
  }
  long tail0 = nvec * V;
  for (long idx = tail0 + blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int c = (int)(idx % C);
    out[idx] = (T)(((float)in[idx] * scale - mean[c]) / std_[c]);
  }
}

// NCHW-contiguous output: iterate output index, gather strided u8 reads
// (u8 gather is absorbed by L2; output writes stay coalesced).
template <typename T>
__global__ void nhwc_pack_nchw_kernel(const u8* __restrict__ in, T* __restrict__ out,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ std_,
                                      float scale, long total, int C, long HW) {
  for (long idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    long n = idx / (C * HW);
    int c = (int)((idx / HW) % C);
    long hw = idx % HW;
    u8 v = in[(n * HW + hw) * C + c];
    out[idx] = (T)(((float)v * scale - mean[c]) / std_[c]);
  }
}

// ---------------------------------------------------------------------------
// Fused flat SGD with momentum:  m = mu*m + g + wd*p;  p -= lr * (m or g+mu*m)
// ---------------------------------------------------------------------------

__global__ void sgd_step_kernel(float* __restrict__ p, const float* __restrict__ g,
                                float* __restrict__ m, float lr, float mu,
                                float wd, int nesterov, long n) {
  constexpr int V = 4;
  const long nvec = n / V;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    f4v pv = *(f4v*)(p + i * V);
    f4v gv = *(const f4v*)(g + i * V);
    f4v mv = *(f4v*)(m + i * V);
    #pragma unroll
    for (int j = 0; j < V; ++j) {
      float grad = gv[j] + wd * pv[j];
      mv[j] = mu * mv[j] + grad;
      float upd = nesterov ? grad + mu * mv[j] : mv[j];
      pv[j] -= lr * upd;
    }
    *(f4v*)(m + i * V) = mv;
    *(f4v*)(p + i * V) = pv;
  }
  for (long i = nvec * V + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float grad = g[i] + wd * p[i];
    m[i] = mu * m[i] + grad;
    p[i] -= lr * (nesterov ? grad + mu * m[i] : m[i]);
  }
}

extern "C" {

void tfosr_nhwc_pack(const void* in, void* out, const float* mean,
                     const float* std_, float scale, int out_bf16,
                     int channels_last, long total, int C, long HW,
                     hipStream_t s) {
  int grid = tfosr_grid(total / 16, 256);
  if (channels_last) {
    if (out_bf16)
      hipLaunchKernelGGL(nhwc_pack_cl_kernel<bf16_t>, dim3(grid), dim3(256), 0, s,
                         (const u8*)in, (bf16_t*)out, mean, std_, scale, total, C);
    else
      hipLaunchKernelGGL(nhwc_pack_cl_kernel<float>, dim3(grid), dim3(256), 0, s,
                         (const u8*)in, (float*)out, mean, std_, scale, total, C);
  } else {
    grid = tfosr_grid(total, 256);
    if (out_bf16)
      hipLaunchKernelGGL(nhwc_pack_nchw_kernel<bf16_t>, dim3(grid), dim3(256), 0, s,
                         (const u8*)in, (bf16_t*)out, mean, std_, scale, total, C, HW);
    else
      hipLaunchKernelGGL(nhwc_pack_nchw_kernel<float>, dim3(grid), dim3(256), 0, s,
                         (const u8*)in, (float*)out, mean, std_, scale, total, C, HW);
  }
}

void tfosr_sgd_step(float* p, const float* g, float* m, float lr, float mu,
                    float wd, int nesterov, long n, hipStream_t s) {
  int grid = tfosr_grid(n / 4, 256);
  hipLaunchKernelGGL(sgd_step_kernel, dim3(grid), dim3(256), 0, s,
                     p, g, m, lr, mu, wd, nesterov, n);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// Fused flat Adam(W):  m=b1*m+(1-b1)g; v=b2*v+(1-b2)g^2;
//                      p -= lr * mhat/(sqrt(vhat)+eps) (+ decoupled wd)
// ---------------------------------------------------------------------------

__global__ void adam_step_kernel(float* __restrict__ p, const float* __restrict__ g,
                                 float* __restrict__ m, float* __restrict__ v,
                                 float lr, float b1, float b2, float eps,
                                 float wd, float bc1, float bc2, int decoupled,
                                 long n) {
  constexpr int V = 4;
  const long nvec = n / V;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    f4v pv = *(f4v*)(p + i * V);
    f4v gv = *(const f4v*)(g + i * V);
    f4v mv = *(f4v*)(m + i * V);
    f4v vv = *(f4v*)(v + i * V);
    #pragma unroll
    for (int j = 0; j < V; ++j) {
      float grad = decoupled ? gv[j] : gv[j] + wd * pv[j];
      mv[j] = b1 * mv[j] + (1.f - b1) * grad;
      vv[j] = b2 * vv[j] + (1.f - b2) * grad * grad;
      float mhat = mv[j] / bc1;
      float vhat = vv[j] / bc2;
      float upd = mhat / (sqrtf(vhat) + eps);
      if (decoupled) upd += wd * pv[j];
      pv[j] -= lr * upd;
    }
    *(f4v*)(m + i * V) = mv;
    *(f4v*)(v + i * V) = vv;
    *(f4v*)(p + i * V) = pv;
  }
  for (long i = nvec * V + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float grad = decoupled ? g[i] : g[i] + wd * p[i];
    m[i] = b1 * m[i] + (1.f - b1) * grad;
    v[i] = b2 * v[i] + (1.f - b2) * grad * grad;
    float upd = (m[i] / bc1) / (sqrtf(v[i] / bc2) + eps);
    if (decoupled) upd += wd * p[i];
    p[i] -= lr * upd;
  }
}

extern "C" void tfosr_adam_step(float* p, const float* g, float* m, float* v,
                                float lr, float b1, float b2, float eps,
                                float wd, int step, int decoupled, long n,
                                hipStream_t s) {
  int grid = tfosr_grid(n / 4, 256);
  float bc1 = 1.f - powf(b1, (float)step);
  float bc2 = 1.f - powf(b2, (float)step);
  hipLaunchKernelGGL(adam_step_kernel, dim3(grid), dim3(256), 0, s,
                     p, g, m, v, lr, b1, b2, eps, wd, bc1, bc2, decoupled, n);
}

// ---------------------------------------------------------------------------
// Batched weight pack: all per-step weight transforms (cast fp32->bf16,
// permute, flip via negative strides, zero-pad via source-extent clamps) in
// ONE kernel launch. ResNet-50's fused blocks otherwise issue ~300 tiny
// permute/cast kernels per step (~3-4 ms, profiles/README backlog item 3).
//
// Descriptor (ints/longs packed host-side, one per output tensor):
//   dst_off (elems into the bf16 arena), n (elems), od[4] (output dims,
//   innermost last), ss[4] (source strides in ELEMENTS, may be negative),
//   soff (source element offset), sv[4] (source valid extent per dim; an
//   output coordinate >= sv[d] reads 0 — zero padding).
// ---------------------------------------------------------------------------

struct PackDesc {
  const float* src;
  long dst_off;
  long soff;
  int n;
  int od[4];
  int ss[4];
  int sv[4];
};

__global__ void pack_bf16_kernel(const PackDesc* __restrict__ descs,
                                 const long* __restrict__ cum, int ndesc,
                                 bf16_t* __restrict__ arena, long total) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    // binary search for the descriptor owning element i
    int lo = 0, hi = ndesc - 1;
    while (lo < hi) {
      int mid = (lo + hi) >> 1;
      if (i >= cum[mid + 1]) lo = mid + 1; else hi = mid;
    }
    const PackDesc d = descs[lo];
    long r = i - cum[lo];
    int c3 = (int)(r % d.od[3]); r /= d.od[3];
    int c2 = (int)(r % d.od[2]); r /= d.od[2];
    int c1 = (int)(r % d.od[1]); r /= d.od[1];
    int c0 = (int)r;
    float v = 0.f;
    if (c0 < d.sv[0] && c1 < d.sv[1] && c2 < d.sv[2] && c3 < d.sv[3])
      v = d.src[d.soff + (long)c0 * d.ss[0] + (long)c1 * d.ss[1] +
                (long)c2 * d.ss[2] + (long)c3 * d.ss[3]];
    arena[d.dst_off + (i - cum[lo])] = (bf16_t)v;
  }
}

extern "C" void tfosr_pack_bf16(const void* descs, const void* cum, int ndesc,
                                void* arena, long total, hipStream_t s) {
  int threads = 256;
  long want = (total + threads - 1) / threads;
  int blocks = (int)(want > 16384 ? 16384 : (want < 1 ? 1 : want));
  hipLaunchKernelGGL(pack_bf16_kernel, dim3(blocks), dim3(threads), 0, s,
                     (const PackDesc*)descs, (const long*)cum, ndesc,
                     (bf16_t*)arena, total);
}
