// Python bindings for the tfosr gfx950 kernel set. Compiled with hipcc;
// kernels live in the .hip translation units and are reached through the
// extern "C" launchers so torch headers stay out of kernel code.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

extern "C" {
int tfosr_bn_fast_blocks(long, int, int, int);
void tfosr_bn_stats(const void*, int, int, float*, int, int, int, long, hipStream_t);
void tfosr_bn_finalize(const float*, int, float*, float*, float*, float*,
                       long, int, float, float, hipStream_t);
void tfosr_bn_apply(const void*, const void*, void*, unsigned char*,
                    const float*, const float*,
                    const float*, const float*, int, int, int, long, int, long,
                    hipStream_t);
void tfosr_bn_bwd_stats(const void*, const void*, const void*,
                        const unsigned char*, void*,
                        const float*, const float*, float*, int, int, int,
                        int, int, int, long, hipStream_t);
void tfosr_bn_bwd_merge(const float*, int, float*, float*, int, hipStream_t);
void tfosr_bn_bwd_dx(const void*, const void*, const void*,
                     const unsigned char*, const float*,
                     const float*, const float*, const float*, const float*,
                     void*, int, int, int, long, int, long, hipStream_t);
void tfosr_xent_fwd(const void*, const long*, float*, float*, int, int, int,
                    hipStream_t);
void tfosr_xent_bwd(const void*, const float*, const long*, const float*, void*,
                    int, int, int, hipStream_t);
void tfosr_nhwc_pack(const void*, void*, const float*, const float*, float, int,
                     int, long, int, long, hipStream_t);
void tfosr_sgd_step(float*, const float*, float*, float, float, float, int, long,
                    hipStream_t);
void tfosr_adam_step(float*, const float*, float*, float*, float, float, float,
                     float, float, int, int, long, hipStream_t);
void tfosr_gemm_bt(const void*, const void*, void*, int, int, int, int,
                   hipStream_t);
void tfosr_gemm_bt_acc(const void*, const void*, void*, int, int, int, int,
                       int, hipStream_t);
void tfosr_mfma_probe(const short*, const short*, float*, hipStream_t);
void tfosr_maxpool_fwd(const void*, void*, unsigned char*, int, int, int, int,
                       int, int, int, int, int, int, hipStream_t);
void tfosr_maxpool_bwd(const void*, const unsigned char*, void*, int, int, int,
                       int, int, int, int, int, int, int, hipStream_t);
void tfosr_conv3x3(const void*, const void*, const void*, void*, int, int, int,
                   int, int, int, int, int, int, int, hipStream_t);
void tfosr_conv_mfma(const void*, const void*, const void*, void*, int, int,
                     int, int, int, int, int, int, int, int, int, int, int,
                     int, hipStream_t);
void tfosr_conv_wrw(const void*, const void*, float*, int, int, int, int, int,
                    int, int, int, int, int, hipStream_t);
int tfosr_wrw2_split(int, int, long);
void tfosr_conv_wrw2(const void*, const void*, const void*, float*, float*,
                     int, int, int, int, int, int, int, int, int, int, int,
                     int, hipStream_t);
void tfosr_conv_stem(const void*, const void*, const void*, void*, int, int,
                     int, int, int, int, hipStream_t);
void tfosr_conv_par(const void*, const void*, const void*, void*, int, int,
                    int, int, int, int, int, int, int, unsigned long, int,
                    int, int, long, int, hipStream_t);
void tfosr_pack_bf16(const void*, const void*, int, void*, long, hipStream_t);
}

namespace tfosr {
uint32_t crc32c(const uint8_t*, size_t, uint32_t);
struct ScanResult {
  std::string buffer;
  std::vector<std::pair<size_t, size_t>> records;
};
ScanResult scan_file(const std::string&, bool);
void write_file(const std::string&, const std::vector<std::string>&, bool);
}

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

struct BNLayout {
  bool nhwc;
  int N, C;
  long HW;
};

BNLayout bn_layout(const at::Tensor& x) {
  TORCH_CHECK(x.dim() == 4, "expected 4-D NCHW tensor");
  BNLayout l;
  l.N = x.size(0);
  l.C = x.size(1);
  l.HW = (long)x.size(2) * x.size(3);
  if (x.is_contiguous(at::MemoryFormat::ChannelsLast)) {
    l.nhwc = true;
  } else {
    TORCH_CHECK(x.is_contiguous(), "input must be contiguous (NCHW or channels_last)");
    l.nhwc = false;
  }
  return l;
}

int dtype_flag(const at::Tensor& x) {
  if (x.scalar_type() == at::kBFloat16) return 1;
  TORCH_CHECK(x.scalar_type() == at::kFloat, "expected float32 or bfloat16");
  return 0;
}

std::vector<at::Tensor> bn_fwd_train(at::Tensor x, c10::optional<at::Tensor> res,
                                     at::Tensor w, at::Tensor b, at::Tensor rm,
                                     at::Tensor rv, double momentum, double eps,
                                     bool relu) {
  auto l = bn_layout(x);
  int bf = dtype_flag(x);
  const void* res_ptr = nullptr;
  if (res.has_value()) {
    TORCH_CHECK(res->sizes() == x.sizes() && res->scalar_type() == x.scalar_type());
    res_ptr = res->data_ptr();
  }
  auto opts = x.options().dtype(at::kFloat);
  long M = (long)l.N * l.HW;
  int nb = tfosr_bn_fast_blocks(M * l.C, l.C, bf, l.nhwc);
  auto ws = nb > 0 ? at::empty({(long)nb * 2 * l.C}, opts)
                   : at::zeros({2L * l.C}, opts);
  auto save_mean = at::empty({l.C}, opts);
  auto save_rstd = at::empty({l.C}, opts);
  auto y = at::empty_like(x);
  // relu sign bitmask (1 byte per vector) for y-free backward gating
  at::Tensor mask;
  unsigned char* mask_ptr = nullptr;
  if (relu && nb > 0) {
    long nvec = M * l.C / (bf ? 8 : 4);
    mask = at::empty({nvec}, x.options().dtype(at::kByte));
    mask_ptr = mask.data_ptr<unsigned char>();
  } else {
    mask = at::empty({0}, x.options().dtype(at::kByte));
  }
  auto s = cur_stream();
  tfosr_bn_stats(x.data_ptr(), bf, l.nhwc, ws.data_ptr<float>(), nb,
                 l.N, l.C, l.HW, s);
  tfosr_bn_finalize(ws.data_ptr<float>(), nb,
                    save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
                    rm.data_ptr<float>(), rv.data_ptr<float>(), M, l.C,
                    (float)momentum, (float)eps, s);
  tfosr_bn_apply(x.data_ptr(), res_ptr, y.data_ptr(), mask_ptr,
                 save_mean.data_ptr<float>(),
                 save_rstd.data_ptr<float>(), w.data_ptr<float>(),
                 b.data_ptr<float>(), bf, l.nhwc, relu, M * l.C, l.C, l.HW, s);
  return {y, save_mean, save_rstd, mask};
}

at::Tensor bn_fwd_eval(at::Tensor x, c10::optional<at::Tensor> res, at::Tensor w,
                       at::Tensor b, at::Tensor rm, at::Tensor rv, double eps,
                       bool relu) {
  auto l = bn_layout(x);
  int bf = dtype_flag(x);
  const void* res_ptr = res.has_value() ? res->data_ptr() : nullptr;
  auto rstd = at::rsqrt(rv + eps);
  auto y = at::empty_like(x);
  tfosr_bn_apply(x.data_ptr(), res_ptr, y.data_ptr(), nullptr,
                 rm.data_ptr<float>(),
                 rstd.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),
                 bf, l.nhwc, relu, (long)l.N * l.HW * l.C, l.C, l.HW,
                 cur_stream());
  return y;
}

std::vector<at::Tensor> bn_bwd(at::Tensor x, at::Tensor dy, at::Tensor y,
                               at::Tensor mask, at::Tensor w,
                               at::Tensor save_mean,
                               at::Tensor save_rstd, bool relu, bool grad_res) {
  auto l = bn_layout(x);
  int bf = dtype_flag(x);
  dy = l.nhwc ? dy.contiguous(at::MemoryFormat::ChannelsLast) : dy.contiguous();
  auto opts = x.options().dtype(at::kFloat);
  long total = (long)l.N * l.HW * l.C;
  int nb = tfosr_bn_fast_blocks(total, l.C, bf, l.nhwc);
  auto ws = nb > 0 ? at::empty({(long)nb * 2 * l.C}, opts)
                   : at::zeros({2L * l.C}, opts);
  auto dg = at::empty({l.C}, opts);
  auto db = at::empty({l.C}, opts);
  auto dx = at::empty_like(x);
  at::Tensor gout;
  void* gout_ptr = nullptr;
  if (grad_res) {
    gout = at::empty_like(x);
    gout_ptr = gout.data_ptr();
  }
  auto s = cur_stream();
  const unsigned char* mask_ptr =
      mask.numel() ? mask.data_ptr<unsigned char>() : nullptr;
  TORCH_CHECK(!(relu && nb > 0) || mask_ptr != nullptr,
              "fast-path relu backward requires the forward bitmask");
  tfosr_bn_bwd_stats(x.data_ptr(), dy.data_ptr(), y.data_ptr(), mask_ptr,
                     gout_ptr,
                     save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
                     ws.data_ptr<float>(), nb, bf, l.nhwc,
                     relu, l.N, l.C, l.HW, s);
  tfosr_bn_bwd_merge(ws.data_ptr<float>(), nb, dg.data_ptr<float>(),
                     db.data_ptr<float>(), l.C, s);
  if (grad_res) {
    // gout already carries the gated gradient: skip re-gating (and the y read)
    tfosr_bn_bwd_dx(x.data_ptr(), gout.data_ptr(), y.data_ptr(), nullptr,
                    save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
                    w.data_ptr<float>(), dg.data_ptr<float>(), db.data_ptr<float>(),
                    dx.data_ptr(), bf, l.nhwc, /*relu=*/0, total, l.C, l.HW, s);
    return {dx, dg, db, gout};
  }
  tfosr_bn_bwd_dx(x.data_ptr(), dy.data_ptr(), y.data_ptr(), mask_ptr,
                  save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
                  w.data_ptr<float>(), dg.data_ptr<float>(), db.data_ptr<float>(),
                  dx.data_ptr(), bf, l.nhwc, relu, total, l.C, l.HW, s);
  return {dx, dg, db};
}

std::vector<at::Tensor> softmax_xent_fwd(at::Tensor logits, at::Tensor target) {
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(target.scalar_type() == at::kLong);
  int N = logits.size(0), C = logits.size(1);
  auto opts = logits.options().dtype(at::kFloat);
  auto loss = at::empty({N}, opts);
  auto lse = at::empty({N}, opts);
  tfosr_xent_fwd(logits.data_ptr(), target.data_ptr<long>(),
                 loss.data_ptr<float>(), lse.data_ptr<float>(),
                 dtype_flag(logits), N, C, cur_stream());
  return {loss, lse};
}

at::Tensor softmax_xent_bwd(at::Tensor logits, at::Tensor lse, at::Tensor target,
                            at::Tensor gout) {
  int N = logits.size(0), C = logits.size(1);
  auto dlogits = at::empty_like(logits);
  tfosr_xent_bwd(logits.data_ptr(), lse.data_ptr<float>(),
                 target.data_ptr<long>(), gout.contiguous().data_ptr<float>(),
                 dlogits.data_ptr(), dtype_flag(logits), N, C, cur_stream());
  return dlogits;
}

at::Tensor nhwc_pack(at::Tensor x, at::Tensor mean, at::Tensor std_, double scale,
                     bool out_bf16, bool channels_last) {
  TORCH_CHECK(x.dim() == 4 && x.scalar_type() == at::kByte && x.is_contiguous(),
              "nhwc_pack expects contiguous uint8 NHWC");
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  long HW = (long)H * W;
  auto opts = x.options().dtype(out_bf16 ? at::kBFloat16 : at::kFloat);
  at::Tensor out;
  if (channels_last) {
    out = at::empty({N, C, H, W}, opts, at::MemoryFormat::ChannelsLast);
  } else {
    out = at::empty({N, C, H, W}, opts);
  }
  tfosr_nhwc_pack(x.data_ptr(), out.data_ptr(), mean.data_ptr<float>(),
                  std_.data_ptr<float>(), (float)scale, out_bf16, channels_last,
                  (long)N * C * HW, C, HW, cur_stream());
  return out;
}

void sgd_step(at::Tensor p, at::Tensor g, at::Tensor m, double lr, double mu,
              double wd, bool nesterov) {
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous());
  TORCH_CHECK(p.scalar_type() == at::kFloat, "sgd_step expects fp32 params");
  tfosr_sgd_step(p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
                 (float)lr, (float)mu, (float)wd, nesterov, p.numel(),
                 cur_stream());
}

at::Tensor gemm_bt(at::Tensor a, at::Tensor b, bool out_bf16) {
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1),
              "gemm_bt: A[M,K], B[N,K]");
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 && b.scalar_type() == at::kBFloat16);
  int M = a.size(0), N = b.size(0), K = a.size(1);
  if (K % 32 != 0) {
    int Kp = (K + 31) / 32 * 32;
    auto ap = at::zeros({M, Kp}, a.options());
    auto bp = at::zeros({N, Kp}, b.options());
    ap.narrow(1, 0, K).copy_(a);
    bp.narrow(1, 0, K).copy_(b);
    a = ap; b = bp; K = Kp;
  }
  a = a.contiguous();
  b = b.contiguous();
  auto out = at::empty({M, N}, a.options().dtype(out_bf16 ? at::kBFloat16
                                                          : at::kFloat));
  tfosr_gemm_bt(a.data_ptr(), b.data_ptr(), out.data_ptr(), out_bf16, M, N, K,
                cur_stream());
  return out;
}

at::Tensor gemm_bf16(at::Tensor a, at::Tensor b) {
  // A[M,K] @ B[K,N]: feed B^T (contiguous [N,K]) to the kernel
  return gemm_bt(a, b.t().contiguous(), /*out_bf16=*/false);
}

at::Tensor mfma_probe(at::Tensor a, at::Tensor b) {
  TORCH_CHECK(a.sizes() == at::IntArrayRef({64, 8}) &&
              b.sizes() == at::IntArrayRef({64, 8}));
  auto c = at::empty({64, 4}, a.options().dtype(at::kFloat));
  tfosr_mfma_probe(a.contiguous().data_ptr<short>(),
                   b.contiguous().data_ptr<short>(), c.data_ptr<float>(),
                   cur_stream());
  return c;
}

}  // namespace

// bump when the binding surface changes: the Python side refuses to run
// against a stale in-tree .so (clear "rebuild" message instead of a random
// AttributeError mid-training)
#define TFOSR_API_VERSION 4

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("api_version", []() { return TFOSR_API_VERSION; });
  m.def("bn_fwd_train", &bn_fwd_train);
  m.def("bn_fwd_eval", &bn_fwd_eval);
  m.def("bn_bwd", &bn_bwd);
  m.def("softmax_xent_fwd", &softmax_xent_fwd);
  m.def("softmax_xent_bwd", &softmax_xent_bwd);
  m.def("nhwc_pack", &nhwc_pack);
  m.def("sgd_step", &sgd_step);
  m.def("adam_step", [](at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                        double lr, double b1, double b2, double eps, double wd,
                        long step, bool decoupled) {
    TORCH_CHECK(p.scalar_type() == at::kFloat);
    tfosr_adam_step(p.data_ptr<float>(), g.data_ptr<float>(),
                    m.data_ptr<float>(), v.data_ptr<float>(), lr, b1, b2, eps,
                    wd, (int)step, decoupled, p.numel(), cur_stream());
  });
  m.def("gemm_bt", &gemm_bt, py::arg("a"), py::arg("b"), py::arg("out_bf16") = false);
  m.def("gemm_bf16", &gemm_bf16);
  m.def("mfma_probe", &mfma_probe);
  m.def("maxpool_fwd", [](at::Tensor x, long K, long S, long P) {
    TORCH_CHECK(x.dim() == 4 && x.is_contiguous(at::MemoryFormat::ChannelsLast),
                "maxpool_fwd expects channels_last");
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    int OH = (H + 2 * P - K) / S + 1, OW = (W + 2 * P - K) / S + 1;
    TORCH_CHECK(C % (x.scalar_type() == at::kBFloat16 ? 8 : 4) == 0);
    auto y = at::empty({N, C, OH, OW}, x.options(),
                       at::MemoryFormat::ChannelsLast);
    auto idx = at::empty({N, OH, OW, C}, x.options().dtype(at::kByte));
    tfosr_maxpool_fwd(x.data_ptr(), y.data_ptr(), idx.data_ptr<unsigned char>(),
                      x.scalar_type() == at::kBFloat16, N, C, H, W, OH, OW,
                      K, S, P, cur_stream());
    return std::vector<at::Tensor>{y, idx};
  });
  m.def("conv3x3_fwd", [](at::Tensor x, at::Tensor w9, long Cout, long S,
                          long P) {
    TORCH_CHECK(x.dim() == 4 && x.is_contiguous(at::MemoryFormat::ChannelsLast),
                "conv3x3_fwd expects channels_last input");
    TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
                w9.scalar_type() == at::kBFloat16);
    int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
    TORCH_CHECK(Cin % 32 == 0, "conv3x3 requires Cin % 32 == 0");
    TORCH_CHECK(w9.size(1) == 9 * Cin);
    int OH = (H + 2 * P - 3) / S + 1, OW = (W + 2 * P - 3) / S + 1;
    auto y = at::empty({N, Cout, OH, OW}, x.options(),
                       at::MemoryFormat::ChannelsLast);
    auto guard = at::zeros({64}, x.options());
    tfosr_conv3x3(x.data_ptr(), w9.contiguous().data_ptr(), guard.data_ptr(),
                  y.data_ptr(), /*out_bf16=*/1, N, H, W, Cin, Cout, OH, OW,
                  S, P, cur_stream());
    return y;
  });
  // Generalized implicit-GEMM conv: taps in {1,9}, stride S, pad P, input
  // dilation D (the stored input is read as if zero-dilated by D). OH/OW are
  // explicit because backward-data output size is not derivable from the
  // stored input dims. One kernel covers conv1x1/conv3x3 fwd at stride 1/2,
  // their backward-data, and ConvTranspose2d k3 s2 (survey §2.3 rows 1-2).
  m.def("conv_mfma", [](at::Tensor x, at::Tensor wk, long Cout, long fh,
                        long fw, long S, long P, long D, long OH, long OW) {
    TORCH_CHECK(x.dim() == 4 && x.is_contiguous(at::MemoryFormat::ChannelsLast),
                "conv_mfma expects channels_last input");
    TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
                wk.scalar_type() == at::kBFloat16);
    TORCH_CHECK(fh >= 1 && fw >= 1 && fh * fw <= 49, "bad filter dims");
    TORCH_CHECK(D == 1 || D == 2, "input dilation must be 1 or 2");
    long taps = fh * fw;
    int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
    TORCH_CHECK(Cin % 32 == 0, "conv_mfma requires Cin % 32 == 0");
    TORCH_CHECK(wk.size(1) == taps * Cin);
    if (OH < 0) {
      long HV = (H - 1) * D + 1, WV = (W - 1) * D + 1;
      OH = (HV + 2 * P - fh) / S + 1;
      OW = (WV + 2 * P - fw) / S + 1;
    }
    auto y = at::empty({N, Cout, OH, OW}, x.options(),
                       at::MemoryFormat::ChannelsLast);
    auto guard = at::zeros({64}, x.options());
    tfosr_conv_mfma(x.data_ptr(), wk.contiguous().data_ptr(), guard.data_ptr(),
                    y.data_ptr(), /*out_bf16=*/1, N, H, W, Cin, Cout, OH, OW,
                    S, P, taps, fw, D, 0, cur_stream());
    return y;
  });
  // conv_mfma accumulating INTO an existing channels_last tensor:
  // out += conv(x, wk) — fuses the residual-join gradient accumulation the
  // eager autograd would do as a separate whole-tensor add (8.4 ms/step of
  // CUDAFunctor_add at ResNet-50 b1024, profiles/README r01).
  m.def("conv_mfma_acc", [](at::Tensor x, at::Tensor wk, at::Tensor out,
                            long fh, long fw, long S, long P, long D) {
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                out.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
                wk.scalar_type() == at::kBFloat16 &&
                out.scalar_type() == at::kBFloat16);
    int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
    TORCH_CHECK(Cin % 32 == 0, "conv_mfma requires Cin % 32 == 0");
    long taps = fh * fw;
    TORCH_CHECK(wk.size(1) == taps * Cin);
    long Cout = out.size(1), OH = out.size(2), OW = out.size(3);
    TORCH_CHECK(out.size(0) == N);
    auto guard = at::zeros({64}, x.options());
    tfosr_conv_mfma(x.data_ptr(), wk.contiguous().data_ptr(), guard.data_ptr(),
                    out.data_ptr(), /*out_bf16=*/1, N, H, W, Cin, Cout, OH, OW,
                    S, P, taps, fw, D, 1, cur_stream());
    return out;
  });
  // gemm accumulating into an existing [M, N] tensor: C += A @ B^T
  m.def("gemm_bt_acc", [](at::Tensor a, at::Tensor b, at::Tensor c) {
    TORCH_CHECK(a.is_contiguous() && b.is_contiguous() && c.is_contiguous());
    TORCH_CHECK(a.scalar_type() == at::kBFloat16 &&
                b.scalar_type() == at::kBFloat16);
    long M = a.size(0), K = a.size(1), Nn = b.size(0);
    TORCH_CHECK(b.size(1) == K && c.size(0) == M && c.size(1) == Nn);
    TORCH_CHECK(K % 32 == 0);
    tfosr_gemm_bt_acc(a.data_ptr(), b.data_ptr(), c.data_ptr(),
                      c.scalar_type() == at::kBFloat16 ? 1 : 0, M, Nn, K, 1,
                      cur_stream());
    return c;
  });
  // wrw v2: MFMA + hardware transpose-reads + split-M workspace.
  // dy [N,Cout,OH,OW] cl, x [N,Cin,H,W] cl; filter R x S_f, conv stride,
  // pad P. Returns fp32 dW [Cout, R*S_f*Cin].
  m.def("conv_wrw2", [](at::Tensor dy, at::Tensor x, long R, long S_f,
                        long stride, long P) {
    TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                x.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                x.scalar_type() == at::kBFloat16);
    int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
    int Cout = dy.size(1), OH = dy.size(2), OW = dy.size(3);
    TORCH_CHECK(Cin % 8 == 0 && Cout % 8 == 0,
                "conv_wrw2 needs channel counts % 8 == 0");
    long taps = R * S_f;
    TORCH_CHECK(taps == 1 || taps == 9, "conv_wrw2 supports 1x1 and 3x3");
    long M = (long)N * OH * OW;
    int split = tfosr_wrw2_split(Cout, Cin, M);
    long K = taps * (long)Cin;
    auto ws = at::empty({(long)split, (long)Cout * K},
                        x.options().dtype(at::kFloat));
    auto dW = at::empty({(long)Cout, K}, x.options().dtype(at::kFloat));
    auto guard = at::zeros({64}, x.options());
    tfosr_conv_wrw2(dy.data_ptr(), x.data_ptr(), guard.data_ptr(),
                    ws.data_ptr<float>(), dW.data_ptr<float>(), N, H, W, Cin,
                    Cout, OH, OW, R, S_f, stride, P, split, cur_stream());
    return dW;
  });
  // One parity class of a stride-2 backward-data / transposed conv:
  // out[oh*2+oh0, ow*2+ow0] = conv(x, wk_class) with the class's taps
  // (list of (r,s) pairs). wk_class: [Cout, ntaps*Cin]. Writes into `out`
  // (pre-allocated [N, Cout, OHr, OWr] channels_last; classes disjoint).
  // batched weight pack: desc buffer layout must match PackDesc in
  // elementwise.hip (built by ops/packplan.py); one launch packs every
  // per-step weight transform into the bf16 arena
  m.def("pack_bf16", [](at::Tensor descs, at::Tensor cum, long ndesc,
                        at::Tensor arena, long total) {
    TORCH_CHECK(descs.is_cuda() && cum.is_cuda() && arena.is_cuda());
    TORCH_CHECK(arena.scalar_type() == at::kBFloat16);
    tfosr_pack_bf16(descs.data_ptr(), cum.data_ptr(), (int)ndesc,
                    arena.data_ptr(), total, cur_stream());
  });
  m.def("conv_par", [](at::Tensor x, at::Tensor wk, at::Tensor out,
                       std::vector<long> taps_r, std::vector<long> taps_s,
                       long P, bool accum) {
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                out.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
                wk.scalar_type() == at::kBFloat16 &&
                out.scalar_type() == at::kBFloat16);
    int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
    TORCH_CHECK(Cin % 32 == 0);
    long ntaps = (long)taps_r.size();
    TORCH_CHECK(ntaps >= 1 && ntaps <= 8 && taps_s.size() == (size_t)ntaps);
    TORCH_CHECK(wk.size(1) == ntaps * Cin);
    long Cout = out.size(1), OHr = out.size(2), OWr = out.size(3);
    // parity offsets come from the tap list: ihv = (oh*2+oh0) - P + r must be
    // even for every class tap — caller passes oh0/ow0 via taps parity
    // (we derive them from the first tap)
    long oh0 = ((P - taps_r[0]) % 2 + 2) % 2;
    long ow0 = ((P - taps_s[0]) % 2 + 2) % 2;
    long OHs = (OHr - oh0 + 1) / 2, OWs = (OWr - ow0 + 1) / 2;
    if (OHs <= 0 || OWs <= 0) return out;
    unsigned long pack = 0;
    for (long i = 0; i < ntaps; ++i)
      pack |= ((unsigned long)((taps_r[i] << 4) | taps_s[i])) << (i * 8);
    auto guard = at::zeros({64}, x.options());
    tfosr_conv_par(x.data_ptr(), wk.contiguous().data_ptr(), guard.data_ptr(),
                   out.data_ptr(), N, H, W, Cin, (int)Cout, (int)OHs,
                   (int)OWs, (int)P, (int)ntaps, pack, (int)oh0, (int)ow0,
                   (int)OWr, OHr * OWr, accum ? 1 : 0, cur_stream());
    return out;
  });
  // ResNet stem: 7x7/s2/p3 conv over a pre-padded NHWC4 image
  // ([N,4,230,230] channels_last bf16, Cin 3->4 zero-padded, spatial pad 3
  // baked in); weight pre-packed to [Cout, 7*32] with zero columns for the
  // pads. Output [N, Cout, 112, 112].
  m.def("conv_stem", [](at::Tensor x4, at::Tensor w224, long Cout) {
    TORCH_CHECK(x4.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                x4.size(1) == 4, "conv_stem expects NHWC4 padded input");
    TORCH_CHECK(x4.scalar_type() == at::kBFloat16 &&
                w224.scalar_type() == at::kBFloat16);
    TORCH_CHECK(w224.size(1) == 224);
    int N = x4.size(0), Hp = x4.size(2), Wp = x4.size(3);
    int OH = (Hp - 7) / 2 + 1, OW = (Wp - 7) / 2 + 1;
    auto y = at::empty({N, Cout, OH, OW}, x4.options(),
                       at::MemoryFormat::ChannelsLast);
    auto guard = at::zeros({64}, x4.options());
    tfosr_conv_stem(x4.data_ptr(), w224.contiguous().data_ptr(),
                    guard.data_ptr(), y.data_ptr(), N, Hp, Wp, Cout, OH, OW,
                    cur_stream());
    return y;
  });
  // stem weight gradient over the same NHWC4 padded view: returns fp32
  // [Cout, 7*32] (rows x (8 px x 4 ch), pads included — caller unpacks)
  m.def("conv_stem_wrw", [](at::Tensor dy, at::Tensor x4) {
    TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                x4.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                x4.scalar_type() == at::kBFloat16 && x4.size(1) == 4);
    int N = x4.size(0), Hp = x4.size(2), Wp = x4.size(3);
    int Cout = dy.size(1), OH = dy.size(2), OW = dy.size(3);
    long M = (long)N * OH * OW;
    int split = tfosr_wrw2_split(Cout, 32, M);
    long K = 7L * 32;
    auto ws = at::empty({(long)split, (long)Cout * K},
                        x4.options().dtype(at::kFloat));
    auto dW = at::empty({(long)Cout, K}, x4.options().dtype(at::kFloat));
    auto guard = at::zeros({64}, x4.options());
    tfosr_conv_wrw2(dy.data_ptr(), x4.data_ptr(), guard.data_ptr(),
                    ws.data_ptr<float>(), dW.data_ptr<float>(), N, Hp, Wp,
                    /*Cin=*/32, Cout, OH, OW, /*R=*/7, /*S_f=*/1,
                    /*stride=*/2, /*P=*/0, split, cur_stream());
    return dW;
  });
  m.def("conv_wrw", [](at::Tensor dy, at::Tensor x, long R, long S, long P) {
    TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                x.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                x.scalar_type() == at::kBFloat16);
    int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
    int Cout = dy.size(1), OH = dy.size(2), OW = dy.size(3);
    auto dW = at::zeros({Cout, R * S * (long)Cin},
                        x.options().dtype(at::kFloat));
    tfosr_conv_wrw(dy.data_ptr(), x.data_ptr(), dW.data_ptr<float>(), N, H, W,
                   Cin, Cout, OH, OW, R, S, P, cur_stream());
    return dW;
  });
  m.def("maxpool_bwd", [](at::Tensor dy, at::Tensor idx, long H, long W,
                          long K, long S, long P) {
    dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
    int N = dy.size(0), C = dy.size(1), OH = dy.size(2), OW = dy.size(3);
    auto dx = at::empty({N, C, H, W}, dy.options(),
                        at::MemoryFormat::ChannelsLast);
    tfosr_maxpool_bwd(dy.data_ptr(), idx.data_ptr<unsigned char>(),
                      dx.data_ptr(), dy.scalar_type() == at::kBFloat16,
                      N, C, H, W, OH, OW, K, S, P, cur_stream());
    return dx;
  });

  // native TFRecord codec (CPU): bulk scan/write with HW CRC32-C
  m.def("crc32c", [](py::bytes data) {
    std::string s = data;
    return tfosr::crc32c((const uint8_t*)s.data(), s.size(), 0);
  });
  m.def("tfrecord_read_file", [](const std::string& path, bool verify) {
    auto res = tfosr::scan_file(path, verify);
    py::list out;
    for (auto& [off, len] : res.records)
      out.append(py::bytes(res.buffer.data() + off, len));
    return out;
  }, py::arg("path"), py::arg("verify") = false);
  m.def("tfrecord_write_file",
        [](const std::string& path, const std::vector<std::string>& recs,
           bool append) { tfosr::write_file(path, recs, append); },
        py::arg("path"), py::arg("records"), py::arg("append") = false);
}
