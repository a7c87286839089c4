"""Fused-op modules: HIP/CDNA4 kernels on GPU, plain PyTorch on CPU.

Each op the reference delegated to TensorFlow's runtime (survey §2.3) appears
here as a PyTorch-composable module/function whose CUDA-tensor path calls the
hand-written gfx950 kernel from ``csrc/`` and whose CPU path is the fp32
reference implementation the numerics tests compare against.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import get_ext




def _use_wrw2(k, cin, cout):
    """Weight-gradient backend routing. TFOS_WRW: auto (default) routes each
    shape to the measured-faster implementation (MI355X b1024, see
    profiles/README.md — the in-tree transpose-read wrw kernel wins on
    small-channel 3x3 and is within 0.75-1.2x of the library igemm
    elsewhere); mfma2 forces the in-tree kernel, miopen the library."""
    import os
    mode = os.environ.get("TFOS_WRW", "auto")
    if mode == "mfma2":
        return True
    if mode in ("miopen", "mfma"):
        return False
    return k == 3 and cin <= 64 and cout <= 64


# ---------------------------------------------------------------------------
# Fused BatchNorm + ReLU (training fwd/bwd, inference fwd)
# ---------------------------------------------------------------------------

class _FusedBNFn(torch.autograd.Function):
    """BN (+residual add) (+ReLU) with fused CDNA4 kernels.

    relu gating in backward uses sign(y); with the residual variant the gated
    upstream gradient doubles as the residual-branch gradient (one pass)."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, running_mean, running_var,
                momentum, eps, relu):
        ext = get_ext(required=True)
        y, save_mean, save_rstd, mask = ext.bn_fwd_train(
            x, res, weight, bias, running_mean, running_var, momentum, eps, relu)
        ctx.save_for_backward(x, y, mask, weight, save_mean, save_rstd)
        ctx.relu = relu
        ctx.has_res = res is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        x, y, mask, weight, save_mean, save_rstd = ctx.saved_tensors
        out = ext.bn_bwd(x, dy, y, mask, weight, save_mean, save_rstd,
                         ctx.relu, ctx.has_res)
        if ctx.has_res:
            dx, dweight, dbias, dres = out
        else:
            (dx, dweight, dbias), dres = out, None
        return dx, dres, dweight, dbias, None, None, None, None, None


class FusedBNReLU(nn.Module):
    """BatchNorm2d (+ residual add) (+ ReLU) in one HBM pass per stage.

    These ops are HBM-bandwidth-bound on MI355X; fusing normalize, residual
    add and activation into the stat/normalize kernels removes whole-tensor
    round trips vs the unfused composition, and the backward emits the
    residual gradient from the same pass that reduces dgamma/dbeta.
    """

    RELU = True
    ADD = False

    def __init__(self, num_features, eps=1e-5, momentum=0.1):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))

    def forward(self, x, res=None):
        assert (res is not None) == self.ADD, "residual arg mismatch"
        if x.is_cuda:
            # preserve channels_last; otherwise force standard contiguity
            if not x.is_contiguous(memory_format=torch.channels_last):
                x = x.contiguous()
            if res is not None:
                if x.is_contiguous(memory_format=torch.channels_last):
                    res = res.contiguous(memory_format=torch.channels_last)
                else:
                    res = res.contiguous()
            if self.training:
                self.num_batches_tracked += 1
                return _FusedBNFn.apply(
                    x, res, self.weight, self.bias, self.running_mean,
                    self.running_var, self.momentum, self.eps, self.RELU)
            ext = get_ext(required=True)
            if ext is not None:
                return ext.bn_fwd_eval(x, res, self.weight, self.bias,
                                       self.running_mean, self.running_var,
                                       self.eps, self.RELU)
        # CPU / fallback reference path
        y = F.batch_norm(x, self.running_mean, self.running_var, self.weight,
                         self.bias, self.training, self.momentum, self.eps)
        if res is not None:
            y = y + res
        return F.relu(y, inplace=True) if self.RELU else y

    def extra_repr(self):
        return "{}, eps={}, momentum={}, relu={}, add={}".format(
            self.num_features, self.eps, self.momentum, self.RELU, self.ADD)


class FusedBN(FusedBNReLU):
    """Plain fused BatchNorm2d (no activation)."""
    RELU = False
    ADD = False


class FusedBNAddReLU(FusedBNReLU):
    """y = relu(bn(x) + residual) — the ResNet block tail as one op."""
    RELU = True
    ADD = True


# ---------------------------------------------------------------------------
# Fused softmax cross-entropy (sparse labels)
# ---------------------------------------------------------------------------

class _SoftmaxXentFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        ext = get_ext(required=True)
        loss, lse = ext.softmax_xent_fwd(logits, target)
        ctx.save_for_backward(logits, lse, target)
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        ext = get_ext(required=True)
        logits, lse, target = ctx.saved_tensors
        dlogits = ext.softmax_xent_bwd(logits, lse, target, grad_out.contiguous())
        return dlogits, None


def softmax_cross_entropy(logits, target, reduction="mean"):
    """Sparse softmax cross-entropy (reference workloads:
    ``sparse_categorical_crossentropy``, e.g. ``mnist_spark.py:22``,
    ``resnet_cifar_dist.py:210``). Fused single-pass kernel on GPU."""
    if logits.is_cuda and logits.shape[1] >= 64 and get_ext(required=True) is not None:
        loss = _SoftmaxXentFn.apply(logits.contiguous(), target.contiguous())
    else:
        loss = F.cross_entropy(logits.float(), target, reduction="none")
    if reduction == "mean":
        return loss.mean()
    if reduction == "sum":
        return loss.sum()
    return loss


# ---------------------------------------------------------------------------
# NHWC uint8 -> NCHW float/bf16 normalize (the DataFeed->GPU ingest kernel)
# ---------------------------------------------------------------------------

def nhwc_pack(images_u8, mean=None, std=None, out_dtype=torch.bfloat16,
              scale=1.0 / 255, channels_last=False):
    """Decode/pack kernel: NHWC uint8 batch -> normalized NCHW tensor.

    GPU path is one fused kernel (read u8 once, write out_dtype once);
    ``channels_last=True`` keeps the memory order (NHWC *is* channels_last), so
    the pack is a pure vectorized normalize. CPU path is the reference
    composition.
    """
    if images_u8.is_cuda:
        ext = get_ext(required=True)
        if ext is not None:
            m = mean if mean is not None else torch.zeros(
                images_u8.shape[-1], device=images_u8.device)
            s = std if std is not None else torch.ones(
                images_u8.shape[-1], device=images_u8.device)
            return ext.nhwc_pack(images_u8.contiguous(), m.float(), s.float(),
                                 float(scale), out_dtype == torch.bfloat16,
                                 channels_last)
    x = images_u8.to(torch.float32) * scale
    if mean is not None:
        x = x - mean
    if std is not None:
        x = x / std
    x = x.permute(0, 3, 1, 2)
    if channels_last:
        return x.contiguous(memory_format=torch.channels_last).to(out_dtype)
    return x.contiguous().to(out_dtype)


# ---------------------------------------------------------------------------
# Fused flat SGD-with-momentum (one kernel per DDP bucket)
# ---------------------------------------------------------------------------

class BucketSGD:
    """SGD(momentum, weight_decay) operating on DDPEngine flat buckets.

    With flattened params+grads, the whole model updates in a handful of
    kernel launches (one per bucket) instead of one per parameter tensor —
    the launch-bound tail of every small-op optimizer loop disappears.
    """

    def __init__(self, engine, lr=0.1, momentum=0.9, weight_decay=0.0,
                 nesterov=False):
        self.engine = engine
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.nesterov = nesterov
        self._mom = []
        for bucket in engine._buckets:
            self._mom.append(torch.zeros_like(bucket.buffer))

    @torch.no_grad()
    def step(self):
        for bucket, mom in zip(self.engine._buckets, self._mom):
            pf = getattr(bucket, "param_flat", None)
            g = bucket.buffer
            if pf is None:
                # non-flattened params: per-param foreach update
                params = bucket.params
                grads = [bucket.views[p] for p in params]
                moms = []
                off = 0
                for p in params:
                    moms.append(mom[off:off + p.numel()].view_as(p))
                    off += p.numel()
                if self.weight_decay:
                    torch._foreach_add_([g_.view(-1) for g_ in grads],
                                        [p.data.view(-1) for p in params],
                                        alpha=self.weight_decay)
                torch._foreach_mul_(moms, self.momentum)
                torch._foreach_add_(moms, grads)
                upd = moms
                if self.nesterov:
                    upd = torch._foreach_add(grads, moms, alpha=self.momentum)
                torch._foreach_add_([p.data for p in params], upd, alpha=-self.lr)
                continue
            ext = get_ext(required=True) if g.is_cuda else None
            if ext is not None:
                ext.sgd_step(pf, g, mom, self.lr, self.momentum,
                             self.weight_decay, self.nesterov)
            else:
                if self.weight_decay:
                    g = g.add(pf, alpha=self.weight_decay)
                mom.mul_(self.momentum).add_(g)
                upd = g.add(mom, alpha=self.momentum) if self.nesterov else mom
                pf.add_(upd, alpha=-self.lr)

    def zero_grad(self):
        self.engine.zero_grad()


# ---------------------------------------------------------------------------
# Fused NHWC MaxPool2d (argmax saved as a packed window index)
# ---------------------------------------------------------------------------

class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, s, p):
        ext = get_ext(required=True)
        y, idx = ext.maxpool_fwd(x, k, s, p)
        ctx.save_for_backward(idx)
        ctx.params = (x.shape[2], x.shape[3], k, s, p)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        (idx,) = ctx.saved_tensors
        H, W, k, s, p = ctx.params
        dx = ext.maxpool_bwd(dy, idx, H, W, k, s, p)
        return dx, None, None, None


class FusedMaxPool2d(nn.Module):
    """channels_last MaxPool2d; backward is an atomic-free gather over the
    (at most ceil(K/S)^2) windows covering each input pixel."""

    def __init__(self, kernel_size=3, stride=2, padding=1):
        super().__init__()
        self.k, self.s, self.p = kernel_size, stride, padding

    def forward(self, x):
        if x.is_cuda and x.is_contiguous(memory_format=torch.channels_last) \
                and x.shape[1] % (8 if x.dtype == torch.bfloat16 else 4) == 0 \
                and get_ext(required=True) is not None:
            return _MaxPoolFn.apply(x, self.k, self.s, self.p)
        return F.max_pool2d(x, self.k, self.s, self.p)

    def extra_repr(self):
        return "k={}, s={}, p={}".format(self.k, self.s, self.p)


# ---------------------------------------------------------------------------
# 1x1 convolution as an MFMA GEMM (channels_last)
# ---------------------------------------------------------------------------

class _Conv1x1S2Fn(torch.autograd.Function):
    """Stride-2 1x1 conv (ResNet downsample path) on the implicit-GEMM MFMA
    kernel: forward is taps=1/S=2; backward-data is the input-dilated (D=2)
    variant of the same kernel (dx = scatter of dy @ W, computed gather-side);
    weight-grad reduces over the subsampled pixels via the library igemm until
    conv_wrw covers stride 2."""

    @staticmethod
    def forward(ctx, x, weight):
        ext = get_ext(required=True)
        Cout, Cin = weight.shape[0], weight.shape[1]
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        w2d = weight.view(Cout, Cin).to(torch.bfloat16).contiguous()
        y = ext.conv_mfma(x, w2d, Cout, 1, 1, 2, 0, 1, -1, -1)
        ctx.save_for_backward(x, weight)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        x, weight = ctx.saved_tensors
        Cout, Cin = weight.shape[0], weight.shape[1]
        dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        wT = weight.view(Cout, Cin).t().to(torch.bfloat16).contiguous()
        # dx[h,w] = (h,w even) ? dy[h/2,w/2] @ W : 0 — only the even-even
        # parity class has a tap; zero-fill and launch just that class
        dx = torch.zeros(x.shape, device=x.device, dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        _conv_parity(dy, wT.view(Cin, 1, 1, Cout), dx, 1, 0)
        import os
        if Cin % 8 == 0 and Cout % 8 == 0 and _use_wrw2(1, Cin, Cout):
            dw = ext.conv_wrw2(dy, x, 1, 1, 2, 0).view(Cout, Cin, 1, 1)
        else:
            w4 = weight.to(torch.bfloat16).contiguous(
                memory_format=torch.channels_last)
            _, dw, _ = torch.ops.aten.convolution_backward(
                dy, x, w4, None, [2, 2], [0, 0], [1, 1], False, [0, 0], 1,
                [False, True, False])
        return dx, dw.to(weight.dtype)


class _Conv1x1Fn(torch.autograd.Function):
    """Stride-1 1x1 conv on channels_last tensors == GEMM over [N*H*W, Cin].

    forward / input-grad run on the hand-written gfx950 MFMA kernel
    (C = A @ B^T, K-contiguous operands); the weight-grad contraction
    (reduction over the huge M dim) goes through rocBLAS/hipBLASLt — a plain
    library GEMM, per the kernel-usage policy."""

    @staticmethod
    def forward(ctx, x, weight):
        ext = get_ext(required=True)
        N, Cin, H, W = x.shape
        Cout = weight.shape[0]
        # channels_last storage viewed as the [M, Cin] activation matrix
        x2d = x.permute(0, 2, 3, 1).reshape(N * H * W, Cin)
        if x2d.dtype != torch.bfloat16:
            x2d = x2d.to(torch.bfloat16)
        w2d = weight.view(Cout, Cin).to(torch.bfloat16)
        y2d = ext.gemm_bt(x2d, w2d, True)
        ctx.save_for_backward(x2d, w2d)
        ctx.dims = (N, H, W, Cin, Cout, weight.dtype)
        return y2d.view(N, H, W, Cout).permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        x2d, w2d = ctx.saved_tensors
        N, H, W, Cin, Cout, wdtype = ctx.dims
        dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dy2d = dy.permute(0, 2, 3, 1).reshape(N * H * W, Cout)
        # dx[M,Cin] = dy[M,Cout] @ W[Cout,Cin]  ->  gemm_bt(dy, W^T)
        dx2d = ext.gemm_bt(dy2d, w2d.t().contiguous(), True)
        dx = dx2d.view(N, H, W, Cin).permute(0, 3, 1, 2)
        # dW[Cout,Cin] = dy^T @ x: a reduction over the huge M dim.
        # TFOS_WRW: mfma2 (default, transpose-read MFMA kernel), mfma
        # (round-1 TN kernel), miopen (library igemm; hipBLASLt's split-K
        # pick was ~7x slower than either).
        import os
        wrw = os.environ.get("TFOS_WRW", "auto")
        x4d = x2d.view(N, H, W, Cin).permute(0, 3, 1, 2)
        if Cin % 8 == 0 and Cout % 8 == 0 and _use_wrw2(1, Cin, Cout):
            dw = ext.conv_wrw2(dy, x4d, 1, 1, 1, 0).view(Cout, Cin, 1, 1)
        elif wrw == "mfma" and Cin % 8 == 0 and Cout % 8 == 0:
            ext2 = get_ext(required=True)
            dw = ext2.conv_wrw(dy, x4d, 1, 1, 0).view(Cout, Cin, 1, 1)
        else:
            w4d = w2d.view(Cout, Cin, 1, 1)
            _, dw, _ = torch.ops.aten.convolution_backward(
                dy, x4d, w4d, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
                [False, True, False])
        return dx, dw.to(wdtype)


class Conv1x1(nn.Module):
    """Pointwise convolution routed to the MFMA GEMM on GPU (stride 1) or the
    implicit-GEMM conv kernel (stride 2 — the ResNet downsample conv).

    Backend select via TFOS_CONV1X1: 'mfma' (default), 'blas'
    (torch.matmul / hipBLASLt for A/B comparison), 'miopen' (F.conv2d).
    Weight kept in Conv2d's [Cout, Cin, 1, 1] shape for state_dict parity.
    """

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.stride = stride
        self.weight = nn.Parameter(torch.empty(cout, cin, 1, 1))
        nn.init.kaiming_normal_(self.weight, mode="fan_out", nonlinearity="relu")
        # stride-2 backward-data needs K = Cout % 32 == 0 as well
        self._s2_ok = stride == 2 and cin % 32 == 0 and cout % 32 == 0

    def forward(self, x):
        import os
        backend = os.environ.get("TFOS_CONV1X1", "mfma")
        # the MFMA kernel is bf16-only: a silent downcast would corrupt a
        # claimed-fp32 run, so other dtypes use the library conv
        if x.is_cuda and x.dtype == torch.bfloat16 and backend != "miopen":
            x = x.contiguous(memory_format=torch.channels_last)
            if self.stride == 2:
                if self._s2_ok and get_ext(required=True) is not None:
                    return _Conv1x1S2Fn.apply(x, self.weight)
                return F.conv2d(x, self.weight.to(x.dtype), stride=self.stride)
            if backend == "blas":
                N, Cin, H, W = x.shape
                x2d = x.permute(0, 2, 3, 1).reshape(-1, Cin)
                w = self.weight.view(self.weight.shape[0], Cin)
                if x2d.dtype == torch.bfloat16:
                    w = w.to(torch.bfloat16)
                y2d = x2d @ w.t()
                return y2d.view(N, H, W, -1).permute(0, 3, 1, 2)
            if get_ext(required=True) is not None:
                return _Conv1x1Fn.apply(x, self.weight)
        return F.conv2d(x, self.weight.to(x.dtype), stride=self.stride)

    def extra_repr(self):
        return "{}x{} pointwise s{} (MFMA)".format(
            self.weight.shape[1], self.weight.shape[0], self.stride)


# ---------------------------------------------------------------------------
# 3x3 convolution as an implicit MFMA GEMM (channels_last, stride 1)
# ---------------------------------------------------------------------------

class _Conv3x3Fn(torch.autograd.Function):
    """3x3/pad-1 conv (stride 1 or 2) on channels_last bf16 via the
    4-deep-pipelined implicit-GEMM kernel. dgrad = dilated-input conv of dy
    with the flipped/transposed weight (same kernel, D=S); wrw goes through
    MIOpen's tuned igemm by default (TFOS_WRW=mfma for the in-tree kernel)."""

    @staticmethod
    def forward(ctx, x, weight, stride):
        ext = get_ext(required=True)
        Cout, Cin = weight.shape[0], weight.shape[1]
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        w9 = weight.permute(0, 2, 3, 1).reshape(Cout, 9 * Cin)             .to(torch.bfloat16).contiguous()
        y = ext.conv_mfma(x, w9, Cout, 3, 3, stride, 1, 1, -1, -1)
        ctx.save_for_backward(x, weight)
        ctx.stride = stride
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        x, weight = ctx.saved_tensors
        S = ctx.stride
        Cout, Cin = weight.shape[0], weight.shape[1]
        dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        # W'[cin][r][s][cout] = W[cout][2-r][2-s][cin]
        wperm = weight.flip(2, 3).permute(1, 2, 3, 0) \
            .to(torch.bfloat16).contiguous()
        if S == 2:
            dx = torch.empty(x.shape, device=x.device, dtype=torch.bfloat16) \
                .contiguous(memory_format=torch.channels_last)
            _conv_parity(dy, wperm, dx, 3, 1)
        else:
            dx = ext.conv_mfma(dy, wperm.reshape(Cin, 9 * Cout), Cin,
                               3, 3, 1, 1, 1, x.shape[2], x.shape[3])
        import os
        wrw = os.environ.get("TFOS_WRW", "auto")
        if Cin % 8 == 0 and Cout % 8 == 0 and _use_wrw2(3, Cin, Cout):
            # transpose-read MFMA wrw kernel (stride 1 and 2)
            dw9 = ext.conv_wrw2(dy, x, 3, 3, S, 1)
            dw = dw9.view(Cout, 3, 3, Cin).permute(0, 3, 1, 2).contiguous()
        elif wrw == "mfma" and S == 1:
            # round-1 TN kernel (kept for comparison)
            dw9 = ext.conv_wrw(dy, x, 3, 3, 1)
            dw = dw9.view(Cout, 3, 3, Cin).permute(0, 3, 1, 2).contiguous()
        else:
            w4 = weight.to(torch.bfloat16).contiguous(
                memory_format=torch.channels_last)
            _, dw, _ = torch.ops.aten.convolution_backward(
                dy, x, w4, None, [S, S], [1, 1], [1, 1], False, [0, 0], 1,
                [False, True, False])
        return dx, dw.to(weight.dtype), None


class Conv3x3(nn.Module):
    """3x3/pad-1 conv (stride 1 or 2) routed to the implicit-GEMM MFMA kernel.

    Backend select via TFOS_CONV3X3: 'mfma' (default), 'miopen'.
    Weight in Conv2d's [Cout, Cin, 3, 3] shape for state_dict parity.
    """

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.stride = stride
        self.weight = nn.Parameter(torch.empty(cout, cin, 3, 3))
        nn.init.kaiming_normal_(self.weight, mode="fan_out", nonlinearity="relu")
        self._eligible = cin % 32 == 0 and cout % 32 == 0 and cout >= 64 \
            and stride in (1, 2)

    def forward(self, x):
        import os
        backend = os.environ.get("TFOS_CONV3X3", "mfma")
        if x.is_cuda and self._eligible and backend == "mfma"                 and x.dtype == torch.bfloat16                 and get_ext(required=True) is not None:
            x = x.contiguous(memory_format=torch.channels_last)
            return _Conv3x3Fn.apply(x, self.weight, self.stride)
        return F.conv2d(x, self.weight.to(x.dtype), padding=1,
                        stride=self.stride)

    def extra_repr(self):
        return "{}x{} 3x3 s{} (implicit-GEMM MFMA)".format(
            self.weight.shape[1], self.weight.shape[0], self.stride)




def _conv_parity_packed(x, packs, prefix, out, k, pp, accum=False):
    """Parity classes with pre-packed per-class weights (PackPlan views
    named '<prefix>_<ph><pw>' shaped [OC, nr, ns, KC])."""
    ext = get_ext(required=True)
    from .packplan import _class_taps
    for (ph, pw), (rl, sl) in _class_taps(k, pp).items():
        if not rl or not sl or ph >= out.shape[2] or pw >= out.shape[3]:
            continue
        wk = packs["{}_{}{}".format(prefix, ph, pw)]
        oc = wk.shape[-4] if wk.dim() == 4 else wk.shape[0]
        wk2 = wk.reshape(oc, -1)
        taps_r = [r for r in rl for _ in sl]
        taps_s = [s2 for _ in rl for s2 in sl]
        ext.conv_par(x, wk2, out, taps_r, taps_s, pp, accum)
    return out


def _conv_parity(x, wperm, out, k, pp, accum=False):
    """Parity-decomposed conv over a 2x-dilated input: four class launches,
    each with only its valid taps (vs the plain D=2 kernel whose taps miss
    the stored rows 3/4 of the time — 4x wasted MFMA work).

    x: stored (undilated) input [N,KC,h,w] cl bf16; wperm: [OC, k, k, KC]
    direct-conv weight (already flipped/swapped for the backward-data or
    transposed-conv use); out: [N, OC, OH, OW] cl bf16 (every pixel of every
    non-empty class gets written; for k==1 zero-fill `out` first —
    odd-parity classes have no taps)."""
    ext = get_ext(required=True)
    OC = wperm.shape[0]
    KC = wperm.shape[3]
    for ph in (0, 1):
        for pw in (0, 1):
            taps = [(r, s) for r in range(k) for s in range(k)
                    if (ph - pp + r) % 2 == 0 and (pw - pp + s) % 2 == 0]
            if not taps or ph >= out.shape[2] or pw >= out.shape[3]:
                continue
            wk = torch.stack([wperm[:, r, s, :] for r, s in taps], dim=1) \
                .reshape(OC, len(taps) * KC).contiguous()
            ext.conv_par(x, wk, out,
                         [r for r, _ in taps], [s for _, s in taps],
                         pp, accum)
    return out


# ---------------------------------------------------------------------------
# Transposed convolution (U-Net/DeepLab decoders) as a dilated-input conv
# ---------------------------------------------------------------------------

class _ConvT2dFn(torch.autograd.Function):
    """ConvTranspose2d (stride 2) == conv over the zero-dilated input:
    y = conv(x_dil(D=2), W^swap-flip, S=1, P'=k-1-P). Backward-data is the
    plain stride-2 conv of dy with the unswapped weight — both directions run
    on the same implicit-GEMM MFMA kernel (reference workload: pix2pix
    upsample k4 s2 + Conv2DTranspose k3 s2, segmentation_spark.py:85-97)."""

    @staticmethod
    def forward(ctx, x, weight, stride, padding, output_padding):
        ext = get_ext(required=True)
        Cin, Cout, KH, KW = weight.shape
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        # W'[cout][r][s][cin] = W[cin][cout][KH-1-r][KW-1-s]
        wperm = weight.flip(2, 3).permute(1, 2, 3, 0) \
            .to(torch.bfloat16).contiguous()
        H, W = x.shape[2], x.shape[3]
        OH = (H - 1) * stride - 2 * padding + KH + output_padding
        OW = (W - 1) * stride - 2 * padding + KW + output_padding
        y = torch.empty(x.shape[0], Cout, OH, OW, device=x.device,
                        dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        if KH % 2 == 0 and output_padding == 0:
            _conv_parity(x, wperm, y, KH, KH - 1 - padding)
        else:  # odd kernels / output_padding: some classes empty -> zero base
            y.zero_()
            _conv_parity(x, wperm, y, KH, KH - 1 - padding)
        ctx.save_for_backward(x, weight)
        ctx.params = (stride, padding)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        x, weight = ctx.saved_tensors
        S, P = ctx.params
        Cin, Cout, KH, KW = weight.shape
        dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        # dx = plain stride-S conv of dy with W (no flip/swap)
        wd = weight.permute(0, 2, 3, 1).reshape(Cin, KH * KW * Cout)             .to(torch.bfloat16).contiguous()
        dx = ext.conv_mfma(dy, wd, Cin, KH, KW, S, P, 1,
                           x.shape[2], x.shape[3])
        w4 = weight.to(torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        _, dw, _ = torch.ops.aten.convolution_backward(
            dy, x, w4, None, [S, S], [P, P], [1, 1], True, [0, 0], 1,
            [False, True, False])
        return dx, dw.to(weight.dtype), None, None, None


class ConvTranspose2dMFMA(nn.Module):
    """Transposed conv routed to the implicit-GEMM MFMA kernel on GPU.

    Weight kept in nn.ConvTranspose2d's [Cin, Cout, K, K] layout for
    state_dict parity; bias unsupported (decoders here use BN right after).
    """

    def __init__(self, cin, cout, kernel_size=4, stride=2, padding=1,
                 output_padding=0):
        super().__init__()
        self.stride, self.padding = stride, padding
        self.output_padding = output_padding
        self.weight = nn.Parameter(torch.empty(cin, cout, kernel_size,
                                               kernel_size))
        nn.init.kaiming_normal_(self.weight, mode="fan_in", nonlinearity="relu")
        # fwd K = taps*Cin, dgrad K = taps*Cout — both must be 32-multiples
        self._eligible = cin % 32 == 0 and cout % 32 == 0 and stride == 2

    def forward(self, x):
        import os
        # tiny decoders (e.g. 4x4 spatial at the U-Net bottom) under-fill the
        # 256-row implicit-GEMM tiles; those stay on the library conv
        big = x.shape[0] * x.shape[2] * x.shape[3] * 4 >= 32768
        if x.is_cuda and self._eligible and big \
                and x.dtype == torch.bfloat16 \
                and os.environ.get("TFOS_CONVT", "mfma") == "mfma" \
                and get_ext(required=True) is not None:
            x = x.contiguous(memory_format=torch.channels_last)
            return _ConvT2dFn.apply(x, self.weight, self.stride, self.padding,
                                    self.output_padding)
        return F.conv_transpose2d(x, self.weight.to(x.dtype),
                                  stride=self.stride, padding=self.padding,
                                  output_padding=self.output_padding)

    def extra_repr(self):
        return "{}->{} k{} s{} p{} (implicit-GEMM MFMA)".format(
            self.weight.shape[0], self.weight.shape[1], self.weight.shape[2],
            self.stride, self.padding)


# ---------------------------------------------------------------------------
# MFMA GEMM (bf16 inputs, fp32 accumulate) for Dense layers / serving
# ---------------------------------------------------------------------------

def gemm_bf16(a, b):
    """C[m,n] = A[m,k] @ B[k,n] with bf16 inputs, fp32 accumulation.

    GPU path: hand-written MFMA (16x16x32 bf16) LDS-tiled kernel for gfx950.
    CPU path: torch.matmul in fp32.
    """
    if a.is_cuda:
        ext = get_ext(required=True)
        if ext is not None:
            return ext.gemm_bf16(a.contiguous(), b.contiguous())
    return (a.float() @ b.float())


class _GemmBTFn(torch.autograd.Function):
    """Autograd-composable C[M,N] = A[M,K] @ B[N,K]^T on the MFMA kernel.

    dA re-runs the same kernel (dA = dC @ B, K-contiguous after a small
    transpose of B); dB — a contraction over the M dim — uses the library
    GEMM per the kernel-usage policy."""

    @staticmethod
    def forward(ctx, a, b):
        ext = get_ext(required=True)
        a = a.to(torch.bfloat16).contiguous()
        b = b.to(torch.bfloat16).contiguous()
        ctx.save_for_backward(a, b)
        return ext.gemm_bt(a, b, True)

    @staticmethod
    def backward(ctx, dc):
        ext = get_ext(required=True)
        a, b = ctx.saved_tensors
        dc = dc.to(torch.bfloat16).contiguous()
        da = ext.gemm_bt(dc, b.t().contiguous(), True)
        db = (dc.float().t() @ a.float()).to(b.dtype)
        return da, db


class DenseMFMA(nn.Module):
    """Dense/Linear layer on the MFMA GEMM (survey §2.3 GEMM/Dense row:
    reference ``Dense(64, relu)``/``Dense(10)``, ``mnist_spark.py:17-19``).
    Weight/bias layout matches nn.Linear for state_dict parity."""

    def __init__(self, in_features, out_features, bias=True):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        nn.init.kaiming_uniform_(self.weight, a=5 ** 0.5)

    def forward(self, x):
        if x.is_cuda and x.dtype == torch.bfloat16 \
                and self.weight.shape[1] % 32 == 0 \
                and get_ext(required=True) is not None:
            y = _GemmBTFn.apply(x, self.weight)
            if self.bias is not None:
                y = y + self.bias.to(y.dtype)
            return y
        return F.linear(x, self.weight.to(x.dtype),
                        None if self.bias is None else self.bias.to(x.dtype))

    def __prepare_scriptable__(self):
        # TorchScript export swaps in the equivalent nn.Linear (the custom
        # kernel dispatch isn't scriptable; serving runs the library op)
        m = nn.Linear(self.weight.shape[1], self.weight.shape[0],
                      bias=self.bias is not None)
        m.weight = self.weight
        if self.bias is not None:
            m.bias = self.bias
        m.train(self.training)
        return m


class Conv2dIm2colMFMA(nn.Module):
    """Small-Cin conv (e.g. MNIST's Conv2D(1->32, k3)) via im2col into the
    MFMA GEMM, K zero-padded to a 32-multiple. The unfold/pad are tiny at
    MNIST scale; the matmul — the hot part — runs on the in-tree kernel.
    Weight/bias layout matches nn.Conv2d."""

    def __init__(self, cin, cout, kernel_size, bias=True):
        super().__init__()
        self.k = kernel_size
        self.weight = nn.Parameter(torch.empty(cout, cin, kernel_size,
                                               kernel_size))
        self.bias = nn.Parameter(torch.zeros(cout)) if bias else None
        nn.init.kaiming_uniform_(self.weight, a=5 ** 0.5)

    def forward(self, x):
        if x.is_cuda and x.dtype == torch.bfloat16 \
                and get_ext(required=True) is not None:
            N, Cin, H, W = x.shape
            Cout = self.weight.shape[0]
            OH, OW = H - self.k + 1, W - self.k + 1
            K = Cin * self.k * self.k
            Kpad = (K + 31) // 32 * 32
            cols = F.unfold(x, self.k)                    # [N, K, L]
            a = cols.transpose(1, 2).reshape(-1, K)       # [N*L, K]
            a = F.pad(a, (0, Kpad - K))
            w2 = F.pad(self.weight.view(Cout, K), (0, Kpad - K))
            y2 = _GemmBTFn.apply(a, w2)                   # [N*L, Cout]
            if self.bias is not None:
                y2 = y2 + self.bias.to(y2.dtype)
            return y2.view(N, OH * OW, Cout).transpose(1, 2)                 .reshape(N, Cout, OH, OW)
        b = None if self.bias is None else self.bias.to(x.dtype)
        return F.conv2d(x, self.weight.to(x.dtype), b)

    def __prepare_scriptable__(self):
        m = nn.Conv2d(self.weight.shape[1], self.weight.shape[0], self.k,
                      bias=self.bias is not None)
        m.weight = self.weight
        if self.bias is not None:
            m.bias = self.bias
        m.train(self.training)
        return m


class BucketAdam:
    """Adam(W) over DDPEngine flat buckets — one fused kernel per bucket on
    GPU (fp32 m/v state), torch ops on CPU. ``decoupled=True`` = AdamW."""

    def __init__(self, engine, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, decoupled=False):
        self.engine = engine
        self.lr = lr
        self.b1, self.b2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.decoupled = decoupled
        self.step_count = 0
        self._m = [torch.zeros_like(b.buffer) for b in engine._buckets]
        self._v = [torch.zeros_like(b.buffer) for b in engine._buckets]

    @torch.no_grad()
    def step(self):
        self.step_count += 1
        for bucket, m, v in zip(self.engine._buckets, self._m, self._v):
            pf = bucket.param_flat
            g = bucket.buffer
            assert pf is not None, "BucketAdam requires flatten_params=True"
            ext = get_ext(required=True) if g.is_cuda else None
            if ext is not None:
                ext.adam_step(pf, g, m, v, self.lr, self.b1, self.b2, self.eps,
                              self.weight_decay, self.step_count, self.decoupled)
                continue
            grad = g if self.decoupled or not self.weight_decay \
                else g.add(pf, alpha=self.weight_decay)
            m.mul_(self.b1).add_(grad, alpha=1 - self.b1)
            v.mul_(self.b2).addcmul_(grad, grad, value=1 - self.b2)
            bc1 = 1 - self.b1 ** self.step_count
            bc2 = 1 - self.b2 ** self.step_count
            upd = (m / bc1) / ((v / bc2).sqrt() + self.eps)
            if self.decoupled and self.weight_decay:
                upd = upd + self.weight_decay * pf
            pf.add_(upd, alpha=-self.lr)

    def zero_grad(self):
        self.engine.zero_grad()


# ---------------------------------------------------------------------------
# Fully-fused ResNet Bottleneck (manual backward, zero autograd glue)
# ---------------------------------------------------------------------------

class _BottleneckFn(torch.autograd.Function):
    """One autograd node for the whole bottleneck block.

    Eager autograd accumulates the residual-join gradients (the block input
    feeds both conv1 and the identity/downsample path) with whole-tensor adds
    — 8.4 ms/step of CUDAFunctor_add at ResNet-50 b1024 (profiles/README
    r01). The manual backward instead makes the conv1 (and downsample)
    backward-data kernels accumulate straight into the residual gradient
    buffer (gemm_bt_acc / conv_mfma_acc epilogues), so no separate add runs.

    Tensors bf16 channels_last; BN stats/params fp32. Training mode only;
    saved set (t*, a*, masks, stats) matches what the unfused composition
    saves, so peak memory is unchanged.
    """

    @staticmethod
    def forward(ctx, x, w1, g1, b1, rm1, rv1, w2, g2, b2, rm2, rv2,
                w3, g3, b3, rm3, rv3, wd, gd, bd, rmd, rvd,
                stride, momentum, eps, packs=None):
        ext = get_ext(required=True)
        N, Cin, H, W = x.shape
        C1, C2, C3 = w1.shape[0], w2.shape[0], w3.shape[0]

        def as2d(t):
            return t.permute(0, 2, 3, 1).reshape(-1, t.shape[1])

        def as4d(t2, h, w):
            return t2.view(N, h, w, -1).permute(0, 3, 1, 2)

        # packed bf16 weights: from the batched PackPlan when supplied
        # (one kernel for the whole model per optimizer step), else inline
        if packs is not None:
            w1b = packs["w1b"].reshape(C1, Cin)
            w9 = packs["w9"].reshape(C2, 9 * C1)
            w3b = packs["w3b"].reshape(C3, C2)
        else:
            w1b = w1.view(C1, Cin).to(torch.bfloat16).contiguous()
            w9 = w2.permute(0, 2, 3, 1).reshape(C2, 9 * C1) \
                .to(torch.bfloat16).contiguous()
            w3b = w3.view(C3, C2).to(torch.bfloat16).contiguous()
        t1 = as4d(ext.gemm_bt(as2d(x), w1b, True), H, W)
        a1, m1, r1, k1 = ext.bn_fwd_train(t1, None, g1, b1, rm1, rv1,
                                          momentum, eps, True)
        t2 = ext.conv_mfma(a1, w9, C2, 3, 3, stride, 1, 1, -1, -1)
        a2, m2, r2, k2 = ext.bn_fwd_train(t2, None, g2, b2, rm2, rv2,
                                          momentum, eps, True)
        OH, OW = t2.shape[2], t2.shape[3]
        t3 = as4d(ext.gemm_bt(as2d(a2), w3b, True), OH, OW)

        if wd is not None:
            wdb = packs["wdb"].reshape(C3, Cin) if packs is not None \
                else wd.view(C3, Cin).to(torch.bfloat16).contiguous()
            if stride == 1:
                td = as4d(ext.gemm_bt(as2d(x), wdb, True), H, W)
            else:
                td = ext.conv_mfma(x, wdb, C3, 1, 1, stride, 0, 1, -1, -1)
            idn, md, rd, _kd = ext.bn_fwd_train(td, None, gd, bd, rmd, rvd,
                                                momentum, eps, False)
            opt = [td, idn, md, rd]
        else:
            idn = None
            opt = []
        res = idn if idn is not None else x
        y, m3, r3, k3 = ext.bn_fwd_train(t3, res, g3, b3, rm3, rv3,
                                         momentum, eps, True)

        ctx.save_for_backward(x, w1, g1, t1, a1, m1, r1, k1,
                              w2, g2, t2, a2, m2, r2, k2,
                              w3, g3, t3, m3, r3, k3, y, wd, gd, *opt)
        ctx.meta = (stride, wd is not None)
        ctx.packs = packs
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        (x, w1, g1, t1, a1, m1, r1, k1, w2, g2, t2, a2, m2, r2, k2,
         w3, g3, t3, m3, r3, k3, y, wd, gd, *opt) = ctx.saved_tensors
        stride, has_down = ctx.meta
        packs = ctx.packs
        N, Cin, H, W = x.shape
        C1, C2, C3 = w1.shape[0], w2.shape[0], w3.shape[0]
        OH, OW = t2.shape[2], t2.shape[3]
        dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)

        def as2d(t):
            return t.permute(0, 2, 3, 1).reshape(-1, t.shape[1])

        def as4d(t2d, h, w):
            return t2d.view(N, h, w, -1).permute(0, 3, 1, 2)

        def cl(t):
            return t.contiguous(memory_format=torch.channels_last)

        # block tail: bn3(+res+relu) backward -> dt3 plus the gated residual
        # gradient dres (doubles as the dx accumulation buffer below)
        dt3, dg3, db3, dres = ext.bn_bwd(t3, dy, y, k3, g3, m3, r3,
                                         True, True)
        dt3 = cl(dt3)

        def lib_wrw(dyt, xt, w4, s_, p_):
            w4 = w4.to(torch.bfloat16).contiguous(
                memory_format=torch.channels_last)
            _, dw, _ = torch.ops.aten.convolution_backward(
                dyt, xt, w4, None, [s_, s_], [p_, p_], [1, 1], False, [0, 0],
                1, [False, True, False])
            return dw

        # conv3 (1x1): dgrad + wrw
        w3bT = packs["w3bT"].reshape(C2, C3) if packs is not None \
            else w3.view(C3, C2).to(torch.bfloat16).t().contiguous()
        da2 = cl(as4d(ext.gemm_bt(as2d(dt3), w3bT, True), OH, OW))
        dw3 = ext.conv_wrw2(dt3, a2, 1, 1, 1, 0).view(C3, C2, 1, 1) \
            if _use_wrw2(1, C2, C3) \
            else lib_wrw(dt3, a2, w3.view(C3, C2, 1, 1), 1, 0)

        # bn2+relu backward
        dt2, dg2, db2 = ext.bn_bwd(t2, da2, a2, k2, g2, m2, r2, True, False)
        dt2 = cl(dt2)

        # conv2 (3x3, stride s): dgrad (parity-decomposed for s2) + wrw
        if stride == 2:
            da1 = torch.empty(N, C1, H, W, device=x.device,
                              dtype=torch.bfloat16) \
                .contiguous(memory_format=torch.channels_last)
            if packs is not None:
                _conv_parity_packed(dt2, packs, "w9p", da1, 3, 1)
            else:
                wperm2 = w2.flip(2, 3).permute(1, 2, 3, 0) \
                    .to(torch.bfloat16).contiguous()
                _conv_parity(dt2, wperm2, da1, 3, 1)
        else:
            w9p = packs["w9p"].reshape(C1, 9 * C2) if packs is not None \
                else w2.flip(2, 3).permute(1, 2, 3, 0).reshape(C1, 9 * C2) \
                .to(torch.bfloat16).contiguous()
            da1 = ext.conv_mfma(dt2, w9p, C1, 3, 3, 1, 1, 1, H, W)
        if _use_wrw2(3, C1, C2):
            dw9 = ext.conv_wrw2(dt2, a1, 3, 3, stride, 1)
            dw2 = dw9.view(C2, 3, 3, C1).permute(0, 3, 1, 2).contiguous()
        else:
            dw2 = lib_wrw(dt2, a1, w2, stride, 1)

        # bn1+relu backward
        dt1, dg1, db1 = ext.bn_bwd(t1, da1, a1, k1, g1, m1, r1, True, False)
        dt1 = cl(dt1)

        # conv1 (1x1) wrw
        dw1 = ext.conv_wrw2(dt1, x, 1, 1, 1, 0).view(C1, Cin, 1, 1) \
            if _use_wrw2(1, Cin, C1) \
            else lib_wrw(dt1, x, w1.view(C1, Cin, 1, 1), 1, 0)

        w1bT = packs["w1bT"].reshape(Cin, C1) if packs is not None \
            else w1.view(C1, Cin).to(torch.bfloat16).t().contiguous()
        if has_down:
            td, idn, md, rd = opt
            # downsample path: bnd backward (no relu) then conv dgrad
            dtd, dgd, dbd = ext.bn_bwd(td, dres, idn,
                                       torch.empty(0, dtype=torch.uint8,
                                                   device=td.device),
                                       gd, md, rd, False, False)
            dtd = cl(dtd)
            dwd = ext.conv_wrw2(dtd, x, 1, 1, stride, 0) \
                .view(C3, Cin, 1, 1) if _use_wrw2(1, Cin, C3) \
                else lib_wrw(dtd, x, wd.view(C3, Cin, 1, 1), stride, 0)
            # dx = conv1_dgrad, then downsample dgrad ACCUMULATES into it
            dx2d = ext.gemm_bt(as2d(dt1), w1bT, True)
            dx = cl(as4d(dx2d, H, W))
            wdbT = packs["wdbT"].reshape(Cin, C3) if packs is not None \
                else wd.view(C3, Cin).to(torch.bfloat16).t().contiguous()
            if stride == 1:
                ext.gemm_bt_acc(as2d(dtd), wdbT, as2d(dx))
            else:
                # even-even parity class only; odd pixels' contribution is 0
                ext.conv_par(dtd, wdbT, dx, [0], [0], 0, True)
            dwd = dwd.to(wd.dtype)
            dgd_, dbd_ = dgd, dbd
        else:
            # dx = dres + conv1_dgrad — accumulate straight into dres
            dx = dres
            ext.gemm_bt_acc(as2d(dt1), w1bT, as2d(dx))
            dwd = dgd_ = dbd_ = None

        return (dx, dw1.to(w1.dtype), dg1, db1, None, None,
                dw2.to(w2.dtype), dg2, db2, None, None,
                dw3.to(w3.dtype), dg3, db3, None, None,
                dwd, dgd_, dbd_, None, None,
                None, None, None, None)


# ---------------------------------------------------------------------------
# ResNet stem: 7x7/s2/p3 conv on the implicit-GEMM kernel via an NHWC4 view
# ---------------------------------------------------------------------------

class _StemConvFn(torch.autograd.Function):
    """7x7/s2/p3 stem conv (Cin=3) on in-tree kernels.

    The input is packed once to a spatially pre-padded NHWC4 image
    ([N,4,H+6,W+6], channel 3 zero); each filter ROW becomes one uniform
    32-wide K-step (8 px x 4 ch, the 8th px's weight columns are zero), so the
    generic implicit-GEMM kernel runs it with zero guard loads at 1.52x the
    minimal FLOPs (vs 10.7x for a pad-to-32 im2col). The weight gradient runs
    on the same NHWC4 view through the wrw kernel. dgrad is not computed (the
    stem is the first layer; inputs don't carry grad)."""

    @staticmethod
    def forward(ctx, x, weight, w224=None):
        ext = get_ext(required=True)
        N, _, H, W = x.shape
        Cout = weight.shape[0]
        x4 = torch.zeros(N, 4, H + 6, W + 6, dtype=torch.bfloat16,
                         device=x.device) \
            .contiguous(memory_format=torch.channels_last)
        x4[:, :3, 3:H + 3, 3:W + 3] = x
        if w224 is None:
            # w224[cout][r][s*4+c]: s<7, c<3 real else zero
            w224 = torch.zeros(Cout, 7, 8, 4, dtype=torch.bfloat16,
                               device=x.device)
            w224[:, :, :7, :3] = weight.permute(0, 2, 3, 1).to(torch.bfloat16)
        y = ext.conv_stem(x4, w224.reshape(Cout, 224).contiguous(), Cout)
        ctx.save_for_backward(x4, weight)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        x4, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dw224 = ext.conv_stem_wrw(dy, x4)          # [Cout, 224] fp32
        Cout = weight.shape[0]
        dw = dw224.view(Cout, 7, 8, 4)[:, :, :7, :3].permute(0, 3, 1, 2)
        return None, dw.contiguous().to(weight.dtype), None


class StemConv7x7(nn.Module):
    """ResNet-50 stem conv (7x7, stride 2, pad 3, Cin=3) on the MFMA kernels.

    Library fallback on CPU / non-bf16 / when the input requires grad (the
    kernel path does not produce dx — it is the first layer)."""

    def __init__(self, cin=3, cout=64):
        super().__init__()
        assert cin == 3
        self.weight = nn.Parameter(torch.empty(cout, cin, 7, 7))
        nn.init.kaiming_normal_(self.weight, mode="fan_out",
                                nonlinearity="relu")

    def forward(self, x):
        import os
        if x.is_cuda and x.dtype == torch.bfloat16 and not x.requires_grad \
                and os.environ.get("TFOS_STEM", "mfma") == "mfma" \
                and get_ext(required=True) is not None:
            x = x.contiguous(memory_format=torch.channels_last)
            packs = getattr(self, "_tfos_packs", None)
            return _StemConvFn.apply(x, self.weight,
                                     packs["w224"] if packs else None)
        return F.conv2d(x, self.weight.to(x.dtype), stride=2, padding=3)

    def extra_repr(self):
        return "3->{} 7x7 s2 (NHWC4 implicit-GEMM MFMA)".format(
            self.weight.shape[0])


# ---------------------------------------------------------------------------
# TorchScript export support: every fused/MFMA module (and the blocks built
# from them) swaps to an equivalent plain-op module at scripting time, so
# TFNode.export_saved_model produces a loadable TorchScript model for ANY
# in-repo architecture (TFModel.transform / serving / tfosr_infer CLI).
# ---------------------------------------------------------------------------

# torch.jit.script's __prepare_scriptable__ recursion memoizes modules by
# id(); once a replaced original is garbage-collected its id can be recycled
# by a freshly created clone, and the memo then resolves the NEW module to
# the OLD replacement (observed: a stem-conv clone landing where a BatchNorm
# belonged). Keeping every original alive until the export finishes makes id
# reuse impossible. export_saved_model drains this list.
_SCRIPT_CLONE_KEEPALIVE = []


def _bn_clone(src):
    _SCRIPT_CLONE_KEEPALIVE.append(src)
    bn = nn.BatchNorm2d(src.num_features, eps=src.eps, momentum=src.momentum)
    with torch.no_grad():
        bn.weight.copy_(src.weight)
        bn.bias.copy_(src.bias)
        bn.running_mean.copy_(src.running_mean)
        bn.running_var.copy_(src.running_var)
        bn.num_batches_tracked.copy_(src.num_batches_tracked)
    bn.train(src.training)
    return bn


def _conv_clone(m):
    """nn.Conv2d equivalent of Conv1x1 / Conv3x3 / StemConv7x7 (or passthrough
    for an already-plain conv)."""
    if isinstance(m, nn.Conv2d):
        return m
    _SCRIPT_CLONE_KEEPALIVE.append(m)
    k = m.weight.shape[2]
    stride = getattr(m, "stride", 1)
    if isinstance(m, StemConv7x7):
        stride, pad = 2, 3
    else:
        pad = 1 if k == 3 else 0
    new = nn.Conv2d(m.weight.shape[1], m.weight.shape[0], k, stride=stride,
                    padding=pad, bias=False)
    new.weight = m.weight
    new.train(m.training)
    return new


def _convT_clone(m):
    if isinstance(m, nn.ConvTranspose2d):
        return m
    _SCRIPT_CLONE_KEEPALIVE.append(m)
    new = nn.ConvTranspose2d(m.weight.shape[0], m.weight.shape[1],
                             m.weight.shape[2], stride=m.stride,
                             padding=m.padding,
                             output_padding=m.output_padding, bias=False)
    new.weight = m.weight
    new.train(m.training)
    return new


def _fusedbn_prepare(self):
    _SCRIPT_CLONE_KEEPALIVE.append(self)
    bn = _bn_clone(self)
    if self.RELU:
        out = nn.Sequential(bn, nn.ReLU(inplace=True))
        out.train(self.training)
        return out
    return bn


# FusedBNAddReLU takes (x, res) — its parents (the residual blocks) provide
# their own scriptable clones below, so only the one-arg variants get hooks.
def _maxpool_prepare(self):
    _SCRIPT_CLONE_KEEPALIVE.append(self)
    return nn.MaxPool2d(self.k, self.s, self.p)


FusedMaxPool2d.__prepare_scriptable__ = _maxpool_prepare
FusedBNReLU.__prepare_scriptable__ = _fusedbn_prepare
FusedBN.__prepare_scriptable__ = _fusedbn_prepare
Conv1x1.__prepare_scriptable__ = lambda self: _conv_clone(self)
Conv3x3.__prepare_scriptable__ = lambda self: _conv_clone(self)
StemConv7x7.__prepare_scriptable__ = lambda self: _conv_clone(self)
ConvTranspose2dMFMA.__prepare_scriptable__ = lambda self: _convT_clone(self)
