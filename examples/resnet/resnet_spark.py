#!/usr/bin/env python3
"""Distributed ResNet training through the cluster API
(parity: reference ``examples/resnet/resnet_cifar_spark.py`` — the benchmark
workload, with the synthetic-data path of ``resnet_cifar_dist.py:160-168``).

One executor per GPU; InputMode.SPARK feeds synthetic ImageNet/CIFAR-shaped
uint8 rows through the shared-memory ring; workers run sync-SGD with bucketed
RCCL all-reduce overlapped with backward (gloo on CPU).

  python examples/resnet/resnet_spark.py --model resnet56_cifar \
      --cluster_size 2 --steps 20 --num_gpus 0     # CPU smoke
  python examples/resnet/resnet_spark.py --model resnet50 --cluster_size 8
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def map_fun(args, ctx):
    import numpy as np
    import torch

    from tensorflowonspark_amd.models import resnet50, resnet56_cifar
    from tensorflowonspark_amd.ops.modules import (BucketSGD, nhwc_pack,
                                                   softmax_cross_entropy)
    from tensorflowonspark_amd.parallel import DDPEngine
    from tensorflowonspark_amd.utils import checkpoint as ckpt
    from tensorflowonspark_amd.utils.metrics import StepTimer

    ctx.init_process_group()
    device = ctx.device
    use_cuda = device.type == "cuda"
    if args.model == "resnet50":
        model, shape, ncls = resnet50().to(device), (224, 224, 3), 1000
    else:
        model, shape, ncls = resnet56_cifar().to(device), (32, 32, 3), 10
    if use_cuda:
        model = model.to(memory_format=torch.channels_last)
    model.train()
    engine = DDPEngine(model)
    opt = BucketSGD(engine, lr=0.1 * ctx.world_size * args.batch_size / 256,
                    momentum=0.9, weight_decay=1e-4)
    feed = ctx.get_data_feed(train_mode=True)
    timer = StepTimer(args.batch_size * ctx.world_size, log_every=10)
    amp = torch.autocast(device.type, dtype=torch.bfloat16, enabled=use_cuda)

    step = 0
    while True:
        batch = feed.next_batch(args.batch_size) if not feed.should_stop() else []
        if not engine.all_ranks_ready(len(batch) > 0):
            break
        imgs = np.asarray([r[0] for r in batch], dtype=np.uint8).reshape(
            (-1,) + shape)
        labs = np.asarray([r[1] for r in batch], dtype=np.int64)
        x_u8 = torch.as_tensor(imgs, device=device)
        y = torch.as_tensor(labs, device=device)
        opt.zero_grad()
        with amp:
            x = nhwc_pack(x_u8, out_dtype=torch.bfloat16 if use_cuda
                          else torch.float32, channels_last=use_cuda)
            loss = softmax_cross_entropy(model(x), y)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        timer.step()
        step += 1
    feed.terminate()
    if ctx.is_chief:
        ckpt.save_checkpoint(args.model_dir, step, model)
    torch.distributed.destroy_process_group()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet56_cifar",
                   choices=["resnet50", "resnet56_cifar"])
    p.add_argument("--cluster_size", type=int, default=2)
    p.add_argument("--batch_size", type=int, default=32)
    p.add_argument("--steps", type=int, default=20,
                   help="synthetic rows fed = steps * batch * cluster_size")
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--num_gpus", type=int, default=1)
    p.add_argument("--model_dir", default="resnet_model")
    args = p.parse_args()
    # executor working dirs differ from the driver's: path args
    # must be absolute (shared-filesystem semantics, as on a real
    # cluster)
    args.model_dir = os.path.abspath(args.model_dir)

    import numpy as np

    from tensorflowonspark_amd import TFCluster
    from tensorflowonspark_amd.local_context import LocalSparkContext
    sc = LocalSparkContext(num_executors=args.cluster_size)

    shape = (224, 224, 3) if args.model == "resnet50" else (32, 32, 3)
    ncls = 1000 if args.model == "resnet50" else 10
    rng = np.random.default_rng(0)
    n = args.steps * args.batch_size * args.cluster_size
    rows = [(rng.integers(0, 256, size=shape, dtype=np.uint8).reshape(-1),
             int(rng.integers(0, ncls))) for _ in range(n)]

    cluster = TFCluster.run(sc, map_fun, args, args.cluster_size,
                            input_mode=TFCluster.InputMode.SPARK,
                            master_node="chief", num_gpus=args.num_gpus,
                            ring_slot_bytes=max(8 << 20,
                                                args.batch_size * 224 * 224 * 3 * 2))
    cluster.train(sc.parallelize(rows, args.cluster_size * 2), args.epochs)
    cluster.shutdown(grace_secs=5)
    sc.stop()
    print("done")


if __name__ == "__main__":
    main()
