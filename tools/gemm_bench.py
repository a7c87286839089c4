#!/usr/bin/env python3
"""A/B microbenchmark: tfosr gemm_bt kernels vs hipBLASLt (torch.matmul)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tensorflowonspark_amd.ops import get_ext  # noqa: E402


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def main():
    ext = get_ext(required=True)
    shapes = [(4096, 4096, 4096), (8192, 8192, 8192),
              (100352, 256, 512), (100352, 512, 256), (25088, 1024, 256)]
    for M, N, K in shapes:
        a = (torch.randn(M, K, device="cuda") / 8).bfloat16()
        b = (torch.randn(N, K, device="cuda") / 8).bfloat16()
        flops = 2.0 * M * N * K
        # correctness spot check vs library GEMM
        c = ext.gemm_bt(a[:512], b[:512], True).float()
        ref = (a[:512].float() @ b[:512].float().t())
        err = (c - ref).abs().max().item()
        t_ours = bench(lambda: ext.gemm_bt(a, b, True))
        t_blas = bench(lambda: a @ b.t())
        print("M={} N={} K={}: ours {:7.1f} TF  blas {:7.1f} TF  maxerr {:.3g}"
              .format(M, N, K, flops / t_ours / 1e12, flops / t_blas / 1e12, err),
              flush=True)


if __name__ == "__main__":
    main()
