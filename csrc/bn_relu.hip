// Fused BatchNorm2d (+residual add)(+ReLU) for gfx950 — training fwd/bwd and
// eval fwd, NHWC (channels_last) fast path + generic fallbacks, fp32/bf16
// activations with fp32 statistics.
//
// These ops are HBM-bound on MI355X; the design rules applied here:
//  * 16 B/lane vector access everywhere (8 bf16 / 4 f32 per thread) — hipcc
//    does not auto-vectorize bf16 loads.
//  * all hot-loop index math in uint32 (64-bit div/mod is ~40 VALU cycles and
//    made the first version 10x off roofline); launchers fall back to the
//    generic kernels when a tensor exceeds 2^31 elements.
//  * NHWC reductions: thread t owns the 8 fixed channels (t*8..t*8+7) mod C
//    (valid when C | 2048 — every ResNet/UNet width), accumulating in
//    registers over a grid-stride loop, one atomicAdd per channel at the end.
//  * fusion: the residual add and ReLU fold into the normalize pass (fwd) and
//    the dgamma/dbeta pass emits the gated upstream gradient (bwd) so the
//    elementwise add/relu/relu' kernels and their whole-tensor round trips
//    disappear.
#include "tfosr_common.h"

typedef unsigned int u32;

// ---------------------------------------------------------------------------
// load/store helpers: VEC elements of T as one 16 B transaction
// ---------------------------------------------------------------------------

template <typename T>
struct VecIO;

template <>
struct VecIO<bf16_t> {
  static constexpr int V = 8;
  using vec_t = s8v;
  __device__ static void load(const bf16_t* p, float* out) {
    vec_t v = *(const vec_t*)p;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      union { short s; bf16_t b; } u;
      u.s = v[j];
      out[j] = (float)u.b;
    }
  }
  __device__ static void store(bf16_t* p, const float* in) {
    vec_t v;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      union { short s; bf16_t b; } u;
      u.b = (bf16_t)in[j];
      v[j] = u.s;
    }
    *(vec_t*)p = v;
  }
};

template <>
struct VecIO<float> {
  static constexpr int V = 4;
  using vec_t = f4v;
  __device__ static void load(const float* p, float* out) {
    f4v v = *(const f4v*)p;
    #pragma unroll
    for (int j = 0; j < 4; ++j) out[j] = v[j];
  }
  __device__ static void store(float* p, const float* in) {
    f4v v;
    #pragma unroll
    for (int j = 0; j < 4; ++j) v[j] = in[j];
    *(f4v*)p = v;
  }
};

// ---------------------------------------------------------------------------
// NHWC fast path — requires C % V == 0 and 2048 % C == 0 (fixed per-thread
// channel slots) and total < 2^31
// ---------------------------------------------------------------------------

// Stage A: per-block partials into workspace[block][2C] via an LDS reduce —
// NO global atomics (a per-thread global-atomic tail serializes ~10^5 adds
// per channel address and was 40x slower than the loads themselves).
template <typename T>
__global__ void stats_nhwc_fast(const T* __restrict__ x,
                                float* __restrict__ partials, u32 nvec, u32 C) {
  extern __shared__ float lds[];  // [2C]: sums then sumsqs
  constexpr int V = VecIO<T>::V;
  const u32 c0 = ((u32)threadIdx.x * V) % C;
  for (u32 i = threadIdx.x; i < 2 * C; i += blockDim.x) lds[i] = 0.f;
  __syncthreads();
  float s[V], q[V], v[V];
  #pragma unroll
  for (int j = 0; j < V; ++j) { s[j] = 0.f; q[j] = 0.f; }
  const u32 stride = gridDim.x * blockDim.x;
  for (u32 i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec; i += stride) {
    VecIO<T>::load(x + (size_t)i * V, v);
    #pragma unroll
    for (int j = 0; j < V; ++j) { s[j] += v[j]; q[j] += v[j] * v[j]; }
  }
  #pragma unroll
  for (int j = 0; j < V; ++j) {
    atomicAdd(&lds[c0 + j], s[j]);
    atomicAdd(&lds[C + c0 + j], q[j]);
  }
  __syncthreads();
  float* out = partials + (size_t)blockIdx.x * 2 * C;
  for (u32 i = threadIdx.x; i < 2 * C; i += blockDim.x) out[i] = lds[i];
}

// Stage B: sum partials over blocks + finalize mean/rstd + running stats.
// One block per channel, 256 threads strided over the partial blocks — a
// single-block C-thread version serialized ~1000 loads per thread (227 us;
// 40x this kernel's data).
__global__ void stats_merge_finalize(const float* __restrict__ partials,
                                     int nblocks, u32 C,
                                     float* __restrict__ save_mean,
                                     float* __restrict__ save_rstd,
                                     float* __restrict__ running_mean,
                                     float* __restrict__ running_var,
                                     long M, float momentum, float eps) {
  __shared__ float scratch[8];
  const u32 c = blockIdx.x;
  float s = 0.f, q = 0.f;
  for (int b = threadIdx.x; b < nblocks; b += blockDim.x) {
    s += partials[(size_t)b * 2 * C + c];
    q += partials[(size_t)b * 2 * C + C + c];
  }
  s = block_sum<256>(s, scratch);
  __syncthreads();
  q = block_sum<256>(q, scratch);
  if (threadIdx.x != 0) return;
  float mean = s / (float)M;
  float var = fmaxf(q / (float)M - mean * mean, 0.f);
  save_mean[c] = mean;
  save_rstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    float unbiased = (M > 1) ? var * (float)M / (float)(M - 1) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// normalize (+add residual)(+relu); with RELU, also emit a per-element
// sign bitmask (1 byte per vector) so backward never re-reads y for gating
template <typename T, bool RELU, bool ADD>
__global__ void bn_apply_fast(const T* __restrict__ x, const T* __restrict__ res,
                              T* __restrict__ y, unsigned char* __restrict__ mask,
                              const float* __restrict__ mean,
                              const float* __restrict__ rstd,
                              const float* __restrict__ w,
                              const float* __restrict__ b, u32 nvec, u32 C) {
  constexpr int V = VecIO<T>::V;
  const u32 c0 = ((u32)threadIdx.x * V) % C;
  float sc[V], sh[V], v[V], r[V];
  #pragma unroll
  for (int j = 0; j < V; ++j) {
    sc[j] = w[c0 + j] * rstd[c0 + j];
    sh[j] = b[c0 + j] - mean[c0 + j] * sc[j];
  }
  const u32 stride = gridDim.x * blockDim.x;
  for (u32 i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec; i += stride) {
    VecIO<T>::load(x + (size_t)i * V, v);
    if (ADD) VecIO<T>::load(res + (size_t)i * V, r);
    unsigned char m = 0;
    #pragma unroll
    for (int j = 0; j < V; ++j) {
      float o = v[j] * sc[j] + sh[j];
      if (ADD) o += r[j];
      if (RELU) {
        if (o > 0.f) m |= (1u << j);
        o = fmaxf(o, 0.f);
      }
      v[j] = o;
    }
    VecIO<T>::store(y + (size_t)i * V, v);
    if (RELU && mask != nullptr) mask[i] = m;
  }
}

// backward reductions: g = dy * gate(y); dbeta += g; dgamma += g * xhat;
// optionally write g out (the residual branch gradient for the ADD variant)
template <typename T, bool RELU, bool WRITE_G>
__global__ void bwd_stats_fast(const T* __restrict__ x, const T* __restrict__ dy,
                               const unsigned char* __restrict__ mask,
                               T* __restrict__ gout,
                               const float* __restrict__ mean,
                               const float* __restrict__ rstd,
                               float* __restrict__ partials, u32 nvec, u32 C) {
  extern __shared__ float lds[];  // [2C]: dgamma then dbeta
  constexpr int V = VecIO<T>::V;
  const u32 c0 = ((u32)threadIdx.x * V) % C;
  for (u32 i = threadIdx.x; i < 2 * C; i += blockDim.x) lds[i] = 0.f;
  __syncthreads();
  float sg[V], sb[V], xv[V], dv[V];
  float mu[V], rs[V];
  #pragma unroll
  for (int j = 0; j < V; ++j) {
    sg[j] = 0.f; sb[j] = 0.f;
    mu[j] = mean[c0 + j]; rs[j] = rstd[c0 + j];
  }
  const u32 stride = gridDim.x * blockDim.x;
  for (u32 i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec; i += stride) {
    VecIO<T>::load(x + (size_t)i * V, xv);
    VecIO<T>::load(dy + (size_t)i * V, dv);
    unsigned char m = RELU ? mask[i] : 0;
    #pragma unroll
    for (int j = 0; j < V; ++j) {
      float g = RELU ? (((m >> j) & 1) ? dv[j] : 0.f) : dv[j];
      sb[j] += g;
      sg[j] += g * (xv[j] - mu[j]) * rs[j];
      dv[j] = g;
    }
    if (WRITE_G) VecIO<T>::store(gout + (size_t)i * V, dv);
  }
  #pragma unroll
  for (int j = 0; j < V; ++j) {
    atomicAdd(&lds[c0 + j], sg[j]);
    atomicAdd(&lds[C + c0 + j], sb[j]);
  }
  __syncthreads();
  float* out = partials + (size_t)blockIdx.x * 2 * C;
  for (u32 i = threadIdx.x; i < 2 * C; i += blockDim.x) out[i] = lds[i];
}

__global__ void bwd_stats_merge(const float* __restrict__ partials, int nblocks,
                                u32 C, float* __restrict__ dg,
                                float* __restrict__ db) {
  __shared__ float scratch[8];
  const u32 c = blockIdx.x;
  float sg = 0.f, sb = 0.f;
  for (int b = threadIdx.x; b < nblocks; b += blockDim.x) {
    sg += partials[(size_t)b * 2 * C + c];
    sb += partials[(size_t)b * 2 * C + C + c];
  }
  sg = block_sum<256>(sg, scratch);
  __syncthreads();
  sb = block_sum<256>(sb, scratch);
  if (threadIdx.x == 0) {
    dg[c] = sg;
    db[c] = sb;
  }
}

// dx = w*rstd * (g - db/M - xhat * dg/M); g recomputed from (dy, y) or read
// from the gated gradient written by bwd_stats (GATED=false).
template <typename T, bool RELU>
__global__ void bwd_dx_fast(const T* __restrict__ x, const T* __restrict__ dy,
                            const unsigned char* __restrict__ mask,
                            T* __restrict__ dx,
                            const float* __restrict__ mean,
                            const float* __restrict__ rstd,
                            const float* __restrict__ w,
                            const float* __restrict__ dg,
                            const float* __restrict__ db,
                            u32 nvec, u32 C, float invM) {
  constexpr int V = VecIO<T>::V;
  const u32 c0 = ((u32)threadIdx.x * V) % C;
  float sc[V], mu[V], rs[V], dgm[V], dbm[V], xv[V], dv[V];
  #pragma unroll
  for (int j = 0; j < V; ++j) {
    mu[j] = mean[c0 + j]; rs[j] = rstd[c0 + j];
    sc[j] = w[c0 + j] * rs[j];
    dgm[j] = dg[c0 + j] * invM;
    dbm[j] = db[c0 + j] * invM;
  }
  const u32 stride = gridDim.x * blockDim.x;
  for (u32 i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec; i += stride) {
    VecIO<T>::load(x + (size_t)i * V, xv);
    VecIO<T>::load(dy + (size_t)i * V, dv);
    unsigned char m = RELU ? mask[i] : 0;
    #pragma unroll
    for (int j = 0; j < V; ++j) {
      float g = RELU ? (((m >> j) & 1) ? dv[j] : 0.f) : dv[j];
      float xhat = (xv[j] - mu[j]) * rs[j];
      dv[j] = sc[j] * (g - dbm[j] - xhat * dgm[j]);
    }
    VecIO<T>::store(dx + (size_t)i * V, dv);
  }
}

// ---------------------------------------------------------------------------
// Generic fallbacks (any layout/shape): scalar, 64-bit safe
// ---------------------------------------------------------------------------

template <typename T, bool NHWC>
__global__ void stats_generic(const T* __restrict__ x, float* __restrict__ ws,
                              long total, int C, long HW) {
  // atomic per element — this path only serves odd shapes/layouts
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int c = NHWC ? (int)(idx % C) : (int)((idx / HW) % C);
    float v = (float)x[idx];
    atomicAdd(&ws[c], v);
    atomicAdd(&ws[C + c], v * v);
  }
}

template <typename T, bool NHWC, bool RELU, bool ADD>
__global__ void bn_apply_generic(const T* __restrict__ x, const T* __restrict__ res,
                                 T* __restrict__ y,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 const float* __restrict__ w,
                                 const float* __restrict__ b,
                                 long total, int C, long HW) {
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int c = NHWC ? (int)(idx % C) : (int)((idx / HW) % C);
    float o = ((float)x[idx] - mean[c]) * rstd[c] * w[c] + b[c];
    if (ADD) o += (float)res[idx];
    if (RELU) o = fmaxf(o, 0.f);
    y[idx] = (T)o;
  }
}

template <typename T, bool NHWC, bool RELU, bool WRITE_G>
__global__ void bwd_stats_generic(const T* __restrict__ x, const T* __restrict__ dy,
                                  const T* __restrict__ y, T* __restrict__ gout,
                                  const float* __restrict__ mean,
                                  const float* __restrict__ rstd,
                                  float* __restrict__ ws,
                                  long total, int C, long HW) {
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int c = NHWC ? (int)(idx % C) : (int)((idx / HW) % C);
    float g = (float)dy[idx];
    if (RELU) g = ((float)y[idx] > 0.f) ? g : 0.f;
    atomicAdd(&ws[C + c], g);
    atomicAdd(&ws[c], g * ((float)x[idx] - mean[c]) * rstd[c]);
    if (WRITE_G) gout[idx] = (T)g;
  }
}

template <typename T, bool NHWC, bool RELU>
__global__ void bwd_dx_generic(const T* __restrict__ x, const T* __restrict__ dy,
                               const T* __restrict__ y, T* __restrict__ dx,
                               const float* __restrict__ mean,
                               const float* __restrict__ rstd,
                               const float* __restrict__ w,
                               const float* __restrict__ dg,
                               const float* __restrict__ db,
                               long total, int C, long HW, float invM) {
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int c = NHWC ? (int)(idx % C) : (int)((idx / HW) % C);
    float g = (float)dy[idx];
    if (RELU) g = ((float)y[idx] > 0.f) ? g : 0.f;
    float xhat = ((float)x[idx] - mean[c]) * rstd[c];
    dx[idx] = (T)(w[c] * rstd[c] * (g - db[c] * invM - xhat * dg[c] * invM));
  }
}


// ---------------------------------------------------------------------------
// extern "C" launchers — dispatch fast path when eligible
// ---------------------------------------------------------------------------

static inline bool fast_ok(int is_nhwc, long total, int C, int vec) {
  return is_nhwc && total < (1L << 31) && C % vec == 0 && 2048 % C == 0;
}

static inline int fast_grid(long nvec) {
  long g = (nvec + 255) / 256;
  if (g > TFOSR_MAX_GRID) g = TFOSR_MAX_GRID;
  return (int)(g > 0 ? g : 1);
}

static inline int red_grid(long nvec) {
  // reduction stage A: >=32 vectors per thread bounds the partials traffic
  // to ~3% of the tensor; [64, 1024] keeps the chip busy on small layers
  long g = nvec / (256 * 32);
  if (g < 64) g = 64;
  if (g > 1024) g = 1024;
  return (int)g;
}

extern "C" {

// #blocks the two-stage fast reduction will use; 0 => generic path
// (caller sizes the partials workspace as max(nb,1) * 2C floats and must
// ZERO it when nb == 0)
int tfosr_bn_fast_blocks(long total, int C, int is_bf16, int is_nhwc) {
  int vec = is_bf16 ? 8 : 4;
  if (!fast_ok(is_nhwc, total, C, vec)) return 0;
  return red_grid(total / vec);
}

void tfosr_bn_stats(const void* x, int is_bf16, int is_nhwc, float* partials,
                    int nb, int N, int C, long HW, hipStream_t s) {
  const long total = (long)N * HW * C;
  if (nb > 0) {
    int vec = is_bf16 ? 8 : 4;
    u32 nvec = (u32)(total / vec);
    size_t lds = 2 * (size_t)C * sizeof(float);
    if (is_bf16)
      hipLaunchKernelGGL(stats_nhwc_fast<bf16_t>, dim3(nb), dim3(256), lds, s,
                         (const bf16_t*)x, partials, nvec, (u32)C);
    else
      hipLaunchKernelGGL(stats_nhwc_fast<float>, dim3(nb), dim3(256), lds, s,
                         (const float*)x, partials, nvec, (u32)C);
    return;
  }
  int grid = tfosr_grid(total, 256);
#define SG(T, L) hipLaunchKernelGGL((stats_generic<T, L>), dim3(grid), dim3(256), \
                                    0, s, (const T*)x, partials, total, C, HW)
  if (is_bf16) { if (is_nhwc) SG(bf16_t, true); else SG(bf16_t, false); }
  else { if (is_nhwc) SG(float, true); else SG(float, false); }
#undef SG
}

void tfosr_bn_finalize(const float* partials, int nb, float* save_mean,
                       float* save_rstd, float* running_mean, float* running_var,
                       long M, int C, float momentum, float eps, hipStream_t s) {
  hipLaunchKernelGGL(stats_merge_finalize, dim3(C), dim3(256), 0, s,
                     partials, nb > 0 ? nb : 1, (u32)C, save_mean, save_rstd,
                     running_mean, running_var, M, momentum, eps);
}

// relu: 0/1; res: nullptr for plain BN; mask: sign bits out (fast path only)
void tfosr_bn_apply(const void* x, const void* res, void* y, unsigned char* mask,
                    const float* mean,
                    const float* rstd, const float* w, const float* b,
                    int is_bf16, int is_nhwc, int relu, long total, int C,
                    long HW, hipStream_t s) {
  int vec = is_bf16 ? 8 : 4;
  const bool add = res != nullptr;
  if (fast_ok(is_nhwc, total, C, vec)) {
    u32 nvec = (u32)(total / vec);
    int grid = fast_grid(nvec);
#define APPLY_FAST(T, R, A) \
    hipLaunchKernelGGL((bn_apply_fast<T, R, A>), dim3(grid), dim3(256), 0, s, \
                       (const T*)x, (const T*)res, (T*)y, mask, mean, rstd, w, b, \
                       nvec, (u32)C)
    if (is_bf16) {
      if (relu) { if (add) APPLY_FAST(bf16_t, true, true); else APPLY_FAST(bf16_t, true, false); }
      else      { if (add) APPLY_FAST(bf16_t, false, true); else APPLY_FAST(bf16_t, false, false); }
    } else {
      if (relu) { if (add) APPLY_FAST(float, true, true); else APPLY_FAST(float, true, false); }
      else      { if (add) APPLY_FAST(float, false, true); else APPLY_FAST(float, false, false); }
    }
#undef APPLY_FAST
    return;
  }
  int grid = tfosr_grid(total, 256);
#define APPLY_GEN(T, L, R, A) \
  hipLaunchKernelGGL((bn_apply_generic<T, L, R, A>), dim3(grid), dim3(256), 0, s, \
                     (const T*)x, (const T*)res, (T*)y, mean, rstd, w, b, \
                     total, C, HW)
  if (is_bf16) {
    if (is_nhwc) { if (relu) { if (add) APPLY_GEN(bf16_t, true, true, true); else APPLY_GEN(bf16_t, true, true, false); }
                   else { if (add) APPLY_GEN(bf16_t, true, false, true); else APPLY_GEN(bf16_t, true, false, false); } }
    else { if (relu) { if (add) APPLY_GEN(bf16_t, false, true, true); else APPLY_GEN(bf16_t, false, true, false); }
           else { if (add) APPLY_GEN(bf16_t, false, false, true); else APPLY_GEN(bf16_t, false, false, false); } }
  } else {
    if (is_nhwc) { if (relu) { if (add) APPLY_GEN(float, true, true, true); else APPLY_GEN(float, true, true, false); }
                   else { if (add) APPLY_GEN(float, true, false, true); else APPLY_GEN(float, true, false, false); } }
    else { if (relu) { if (add) APPLY_GEN(float, false, true, true); else APPLY_GEN(float, false, true, false); }
           else { if (add) APPLY_GEN(float, false, false, true); else APPLY_GEN(float, false, false, false); } }
  }
#undef APPLY_GEN
}

// gout: non-null => also write gated upstream grad (residual-branch gradient)
// fast path gates from `mask` (written by apply); generic path gates from y
void tfosr_bn_bwd_stats(const void* x, const void* dy, const void* y,
                        const unsigned char* mask, void* gout,
                        const float* mean, const float* rstd, float* partials,
                        int nb, int is_bf16, int is_nhwc, int relu, int N, int C,
                        long HW, hipStream_t s) {
  const long total = (long)N * HW * C;
  const bool wg = gout != nullptr;
  if (nb > 0) {
    int vec = is_bf16 ? 8 : 4;
    u32 nvec = (u32)(total / vec);
    size_t lds = 2 * (size_t)C * sizeof(float);
#define BS_FAST(T, R, W) \
    hipLaunchKernelGGL((bwd_stats_fast<T, R, W>), dim3(nb), dim3(256), lds, s, \
                       (const T*)x, (const T*)dy, mask, (T*)gout, \
                       mean, rstd, partials, nvec, (u32)C)
    if (is_bf16) {
      if (relu) { if (wg) BS_FAST(bf16_t, true, true); else BS_FAST(bf16_t, true, false); }
      else      { if (wg) BS_FAST(bf16_t, false, true); else BS_FAST(bf16_t, false, false); }
    } else {
      if (relu) { if (wg) BS_FAST(float, true, true); else BS_FAST(float, true, false); }
      else      { if (wg) BS_FAST(float, false, true); else BS_FAST(float, false, false); }
    }
#undef BS_FAST
    return;
  }
  int grid = tfosr_grid(total, 256);
#define BS_GEN(T, L, R, W) \
  hipLaunchKernelGGL((bwd_stats_generic<T, L, R, W>), dim3(grid), dim3(256), 0, s, \
                     (const T*)x, (const T*)dy, (const T*)y, (T*)gout, \
                     mean, rstd, partials, total, C, HW)
  if (is_bf16) {
    if (is_nhwc) { if (relu) { if (wg) BS_GEN(bf16_t, true, true, true); else BS_GEN(bf16_t, true, true, false); }
                   else { if (wg) BS_GEN(bf16_t, true, false, true); else BS_GEN(bf16_t, true, false, false); } }
    else { if (relu) { if (wg) BS_GEN(bf16_t, false, true, true); else BS_GEN(bf16_t, false, true, false); }
           else { if (wg) BS_GEN(bf16_t, false, false, true); else BS_GEN(bf16_t, false, false, false); } }
  } else {
    if (is_nhwc) { if (relu) { if (wg) BS_GEN(float, true, true, true); else BS_GEN(float, true, true, false); }
                   else { if (wg) BS_GEN(float, true, false, true); else BS_GEN(float, true, false, false); } }
    else { if (relu) { if (wg) BS_GEN(float, false, true, true); else BS_GEN(float, false, true, false); }
           else { if (wg) BS_GEN(float, false, false, true); else BS_GEN(float, false, false, false); } }
  }
#undef BS_GEN
}

void tfosr_bn_bwd_merge(const float* partials, int nb, float* dg, float* db,
                        int C, hipStream_t s) {
  hipLaunchKernelGGL(bwd_stats_merge, dim3(C), dim3(256), 0, s,
                     partials, nb > 0 ? nb : 1, (u32)C, dg, db);
}

void tfosr_bn_bwd_dx(const void* x, const void* dy, const void* y,
                     const unsigned char* mask,
                     const float* mean, const float* rstd, const float* w,
                     const float* dg, const float* db, void* dx, int is_bf16,
                     int is_nhwc, int relu, long total, int C, long HW,
                     hipStream_t s) {
  const float invM = 1.f / (float)(total / C);
  int vec = is_bf16 ? 8 : 4;
  if (fast_ok(is_nhwc, total, C, vec)) {
    u32 nvec = (u32)(total / vec);
    int grid = fast_grid(nvec);
#define DX_FAST(T, R) \
    hipLaunchKernelGGL((bwd_dx_fast<T, R>), dim3(grid), dim3(256), 0, s, \
                       (const T*)x, (const T*)dy, mask, (T*)dx, \
                       mean, rstd, w, dg, db, nvec, (u32)C, invM)
    if (is_bf16) { if (relu) DX_FAST(bf16_t, true); else DX_FAST(bf16_t, false); }
    else { if (relu) DX_FAST(float, true); else DX_FAST(float, false); }
#undef DX_FAST
    return;
  }
  int grid = tfosr_grid(total, 256);
#define DX_GEN(T, L, R) \
  hipLaunchKernelGGL((bwd_dx_generic<T, L, R>), dim3(grid), dim3(256), 0, s, \
                     (const T*)x, (const T*)dy, (const T*)y, (T*)dx, \
                     mean, rstd, w, dg, db, total, C, HW, invM)
  if (is_bf16) {
    if (is_nhwc) { if (relu) DX_GEN(bf16_t, true, true); else DX_GEN(bf16_t, true, false); }
    else { if (relu) DX_GEN(bf16_t, false, true); else DX_GEN(bf16_t, false, false); }
  } else {
    if (is_nhwc) { if (relu) DX_GEN(float, true, true); else DX_GEN(float, true, false); }
    else { if (relu) DX_GEN(float, false, true); else DX_GEN(float, false, false); }
  }
#undef DX_GEN
}

}  // extern "C"
