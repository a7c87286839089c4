#!/usr/bin/env python3
"""Per-shape conv microbenchmark: in-tree MFMA kernels vs library (MIOpen).

Covers the ResNet-50 stride-2 shapes and the U-Net decoder transposed convs
(VERDICT r01 item 1/7). Run on a GPU box:
    python tools/conv_bench.py [--batch 1024]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def bench_module(mod_fwd, mod_lib, x, tag):
    """Compare fwd and fwd+bwd times of the MFMA module vs library module."""
    xg = x.requires_grad_(True)

    def fwd(m):
        return lambda: m(xg)

    def fwdbwd(m):
        def run():
            if xg.grad is not None:
                xg.grad = None
            y = m(xg)
            y.backward(torch.ones_like(y))
        return run

    f_m = timeit(fwd(mod_fwd))
    f_l = timeit(fwd(mod_lib))
    b_m = timeit(fwdbwd(mod_fwd))
    b_l = timeit(fwdbwd(mod_lib))
    print(f"{tag:45s} fwd {f_m:7.3f} vs lib {f_l:7.3f} ms | "
          f"fwd+bwd {b_m:7.3f} vs lib {b_l:7.3f} ms")
    return f_m, f_l, b_m, b_l


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=1024)
    ap.add_argument("--wrw", action="store_true")
    args = ap.parse_args()
    if args.wrw:
        bench_wrw(args.batch)
        return
    from tensorflowonspark_amd.ops.modules import (Conv1x1, Conv3x3,
                                                   ConvTranspose2dMFMA)
    N = args.batch
    dev = "cuda"
    torch.manual_seed(0)

    # ResNet-50 stride-2 3x3 convs (v1.5: stride on the middle 3x3)
    shapes_3x3s2 = [(128, 56, 128), (256, 28, 256), (512, 14, 512)]
    for cin, hw, cout in shapes_3x3s2:
        x = torch.randn(N, cin, hw, hw, device=dev, dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        m = Conv3x3(cin, cout, stride=2).to(dev)
        lib = torch.nn.Conv2d(cin, cout, 3, 2, 1, bias=False) \
            .to(dev).to(torch.bfloat16).to(memory_format=torch.channels_last)
        bench_module(m, lib, x, f"conv3x3 s2 {cin}->{cout} @{hw}")

    # ResNet-50 downsample 1x1 s2 convs
    shapes_1x1s2 = [(256, 56, 512), (512, 28, 1024), (1024, 14, 2048)]
    for cin, hw, cout in shapes_1x1s2:
        x = torch.randn(N, cin, hw, hw, device=dev, dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        m = Conv1x1(cin, cout, stride=2).to(dev)
        lib = torch.nn.Conv2d(cin, cout, 1, 2, bias=False) \
            .to(dev).to(torch.bfloat16).to(memory_format=torch.channels_last)
        bench_module(m, lib, x, f"conv1x1 s2 {cin}->{cout} @{hw}")

    # U-Net decoder transposed convs (batch scaled down: seg batch is 64)
    Ns = max(1, N // 16)
    for cin, hw, cout in [(320, 4, 512), (608, 8, 256), (288, 16, 128)]:
        x = torch.randn(Ns, cin, hw, hw, device=dev, dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        m = ConvTranspose2dMFMA(cin, cout, 4, 2, 1).to(dev)
        lib = torch.nn.ConvTranspose2d(cin, cout, 4, 2, 1, bias=False) \
            .to(dev).to(torch.bfloat16).to(memory_format=torch.channels_last)
        bench_module(m, lib, x, f"convT k4s2 {cin}->{cout} @{hw} (b{Ns})")


def bench_wrw(batch=1024):
    """wrw per ResNet-50 shape: wrw2 kernel vs MIOpen igemm."""
    import torch
    from tensorflowonspark_amd.ops import get_ext
    ext = get_ext(required=True)
    N = batch
    shapes = [
        # (Cin, HW_in, Cout, k, stride, pad) — ResNet-50 conv zoo
        (64, 56, 64, 3, 1, 1), (128, 28, 128, 3, 1, 1),
        (256, 14, 256, 3, 1, 1), (512, 7, 512, 3, 1, 1),
        (128, 56, 128, 3, 2, 1), (256, 28, 256, 3, 2, 1),
        (512, 14, 512, 3, 2, 1),
        (64, 56, 256, 1, 1, 0), (256, 56, 64, 1, 1, 0),
        (512, 28, 128, 1, 1, 0), (1024, 14, 256, 1, 1, 0),
        (2048, 7, 512, 1, 1, 0), (256, 56, 512, 1, 2, 0),
        (1024, 14, 2048, 1, 2, 0),
    ]
    for cin, hw, cout, k, s, p in shapes:
        oh = (hw + 2 * p - k) // s + 1
        x = torch.randn(N, cin, hw, hw, device="cuda").to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        dy = torch.randn(N, cout, oh, oh, device="cuda").to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        w4 = torch.randn(cout, cin, k, k, device="cuda").to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        t2 = timeit(lambda: ext.conv_wrw2(dy, x, k, k, s, p))
        tm = timeit(lambda: torch.ops.aten.convolution_backward(
            dy, x, w4, None, [s, s], [p, p], [1, 1], False, [0, 0], 1,
            [False, True, False]))
        flops = 2.0 * N * oh * oh * cout * k * k * cin
        print(f"wrw {k}x{k} s{s} {cin:5d}->{cout:5d} @{hw:3d}: "
              f"wrw2 {t2:7.3f} ms ({flops/t2/1e9:6.1f} TF) vs "
              f"MIOpen {tm:7.3f} ms ({flops/tm/1e9:6.1f} TF)")


if __name__ == "__main__":
    main()
