#!/usr/bin/env python3
"""Per-shape BN kernel bandwidth microbenchmark (round-3 backlog item 1).

The aggregate profile says the fused BN set moves ~150-170 GB/step in ~40 ms
(~3.5-4 TB/s effective vs ~8 achievable). This tool measures each BN op
(fwd-train stats+apply, bwd stats+dx) per ResNet-50 layer shape and prints
achieved GB/s, so the slow (shape, kernel) pairs are identifiable exactly.

    python tools/bn_bench.py --batch 1024
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
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=1024)
    args = ap.parse_args()
    from tensorflowonspark_amd.ops import get_ext
    ext = get_ext(required=True)
    N = args.batch
    # ResNet-50 BN shapes: (C, HW) with counts per step implicit
    shapes = [(64, 112), (64, 56), (256, 56), (128, 56), (128, 28),
              (512, 28), (256, 28), (256, 14), (1024, 14), (512, 14),
              (512, 7), (2048, 7)]
    eb = 2  # bf16
    for C, HW in shapes:
        x = torch.randn(N, C, HW, HW, device="cuda").to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        res = torch.randn_like(x)
        dy = torch.randn_like(x)
        g = torch.rand(C, device="cuda") + 0.5
        b = torch.randn(C, device="cuda")
        rm = torch.zeros(C, device="cuda")
        rv = torch.ones(C, device="cuda")
        nelem = x.numel()
        tb = nelem * eb / 1e9  # GB per full tensor pass

        y, m, r, k = ext.bn_fwd_train(x, res, g, b, rm, rv, 0.1, 1e-5, True)

        t_stats = timeit(lambda: ext.bn_fwd_train(x, None, g, b, rm, rv,
                                                  0.1, 1e-5, False))
        t_full = timeit(lambda: ext.bn_fwd_train(x, res, g, b, rm, rv,
                                                 0.1, 1e-5, True))
        t_bwd = timeit(lambda: ext.bn_bwd(x, dy, y, k, g, m, r, True, True))
        # traffic: fwd(no-res) = 2R+1W = 3 passes (+mask ~1/16);
        # fwd(res+relu) = 3R+1W+mask; bwd(res) = stats 2R + gout 1W
        #                + dx 2R+2W (dres) = 7 passes + mask reads
        bw_f = 3 * tb / t_stats
        bw_fr = 4.06 * tb / t_full
        bw_b = 7.12 * tb / t_bwd
        print(f"C={C:5d} HW={HW:4d} ({tb*1000:6.0f} MB/pass): "
              f"fwd {t_stats*1e3:7.3f} ms {bw_f:6.0f} GB/s | "
              f"fwd+res {t_full*1e3:7.3f} ms {bw_fr:6.0f} GB/s | "
              f"bwd {t_bwd*1e3:7.3f} ms {bw_b:6.0f} GB/s")


if __name__ == "__main__":
    main()
