#!/usr/bin/env python3
"""Parallel batch inference from a saved export via TFParallel
(parity: reference ``examples/mnist/keras/mnist_inference.py`` which used
``TFParallel.run`` over a saved_model)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def infer_fn(args, ctx):
    import numpy as np
    import torch

    model = torch.jit.load(os.path.join(args.export_dir, "model.pt"),
                           map_location="cpu")
    if torch.cuda.is_available():
        model = model.cuda()
    model.eval()
    device = next(model.parameters()).device

    rows = []
    with open(args.data) as f:
        for line in f:
            rows.append([int(v) for v in line.strip().split(",")])
    # shard rows across the parallel instances
    shard = rows[ctx.executor_id::args.instances]
    correct = total = 0
    with torch.no_grad():
        for i in range(0, len(shard), 256):
            arr = np.asarray(shard[i:i + 256], dtype=np.float32)
            y = arr[:, 0].astype(np.int64)
            x = torch.as_tensor(arr[:, 1:] / 255.0, device=device).reshape(
                -1, 1, 28, 28)
            pred = model(x).argmax(dim=1).cpu().numpy()
            correct += int((pred == y).sum())
            total += len(y)
    return {"executor": ctx.executor_id, "correct": correct, "total": total}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--instances", type=int, default=2)
    p.add_argument("--data", default="data/mnist/mnist.csv")
    p.add_argument("--export_dir", default="mnist_export")
    args = p.parse_args()
    # executor working dirs differ from the driver's: path args
    # must be absolute (shared-filesystem semantics, as on a real
    # cluster)
    args.data = os.path.abspath(args.data)
    args.export_dir = os.path.abspath(args.export_dir)

    from tensorflowonspark_amd import TFParallel
    from tensorflowonspark_amd.local_context import LocalSparkContext
    sc = LocalSparkContext(num_executors=args.instances)
    results = TFParallel.run(sc, infer_fn, args, args.instances)
    total = sum(r["total"] for r in results)
    correct = sum(r["correct"] for r in results)
    print("accuracy: {}/{} = {:.3f}".format(correct, total, correct / total))
    sc.stop()


if __name__ == "__main__":
    main()
