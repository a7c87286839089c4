#!/usr/bin/env python3
"""TFOS_BUCKET_MB sweep for the DDP all-reduce.

On a multi-GPU node run under torchrun (one rank per GPU, RCCL over xGMI):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \\
        --master-addr 127.0.0.1 tools/bucket_sweep.py --gpus N

On a single GPU it still runs (world 1, all-reduce skipped) and reports the
compute-only baseline, so the sweep harness itself is validated before the
driver's 8-GPU scaling run. Results: one JSON line per bucket size to stdout
and gpurun_out/bucket_sweep.json.

xGMI note (survey §2.4): each MI355X has 7 point-to-point links (~153 GB/s
each); a single ring serializes onto one link, so RCCL needs several in-flight
chunks to approach aggregate bandwidth — bucket size sets that concurrency.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--buckets", default="4,10,25,50,100")
    args = ap.parse_args()

    import torch

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    if world > 1:
        torch.distributed.init_process_group("nccl" if torch.cuda.is_available()
                                             else "gloo")
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))

    from tensorflowonspark_amd.models import resnet50
    from tensorflowonspark_amd.ops.modules import (BucketSGD,
                                                   softmax_cross_entropy)
    from tensorflowonspark_amd.parallel import DDPEngine

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    results = []
    for mb in [float(x) for x in args.buckets.split(",")]:
        torch.manual_seed(0)
        model = resnet50().to(dev).to(memory_format=torch.channels_last)
        model.train()
        engine = DDPEngine(model, bucket_mb=mb)
        opt = BucketSGD(engine, lr=0.05, momentum=0.9)
        x = torch.randn(args.batch, 3, 224, 224, device=dev) \
            .to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
        y = torch.randint(0, 1000, (args.batch,), device=dev)

        def step():
            opt.zero_grad()
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = softmax_cross_entropy(model(x), y)
            loss.backward()
            engine.finalize_backward()
            opt.step()

        for _ in range(args.warmup):
            step()
        if world > 1:
            torch.distributed.barrier()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            step()
        torch.cuda.synchronize()
        if world > 1:
            torch.distributed.barrier()
        dt = (time.perf_counter() - t0) / args.steps
        ips = args.batch * world / dt
        rec = {"bucket_mb": mb, "ms_per_step": round(dt * 1000, 2),
               "images_per_sec": round(ips, 1), "world": world,
               "n_buckets": len(engine._buckets)}
        results.append(rec)
        if rank == 0:
            print(json.dumps(rec), flush=True)
        del model, engine, opt
        torch.cuda.empty_cache()

    if rank == 0:
        os.makedirs("gpurun_out", exist_ok=True)
        with open("gpurun_out/bucket_sweep.json", "w") as f:
            json.dump(results, f, indent=1)
    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
