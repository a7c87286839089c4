#!/usr/bin/env python3
"""U-Net segmentation via InputMode.SPARK
(parity: reference ``examples/segmentation/segmentation_spark.py`` — MobileNetV2
encoder + pix2pix upsample decoder, 128x128 images, 3 mask classes).

Synthetic image/mask pairs stand in for Oxford-IIIT pets (no network)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def map_fun(args, ctx):
    import numpy as np
    import torch

    from tensorflowonspark_amd.models import unet_mobilenet
    from tensorflowonspark_amd.ops.modules import (BucketSGD, nhwc_pack,
                                                   softmax_cross_entropy)
    from tensorflowonspark_amd.parallel import DDPEngine

    ctx.init_process_group()
    device = ctx.device
    use_cuda = device.type == "cuda"
    model = unet_mobilenet(num_classes=3).to(device)
    if use_cuda:
        model = model.to(memory_format=torch.channels_last)
    model.train()
    engine = DDPEngine(model, bucket_mb=8)
    opt = BucketSGD(engine, lr=args.lr, momentum=0.9)
    feed = ctx.get_data_feed(train_mode=True)
    amp = torch.autocast(device.type, dtype=torch.bfloat16, enabled=use_cuda)

    while True:
        batch = feed.next_batch(args.batch_size) if not feed.should_stop() else []
        if not engine.all_ranks_ready(len(batch) > 0):
            break
        imgs = np.asarray([r[0] for r in batch], dtype=np.uint8).reshape(
            -1, 128, 128, 3)
        masks = np.asarray([r[1] for r in batch], dtype=np.int64).reshape(
            -1, 128, 128)
        x_u8 = torch.as_tensor(imgs, device=device)
        y = torch.as_tensor(masks, device=device)
        opt.zero_grad()
        with amp:
            x = nhwc_pack(x_u8, out_dtype=torch.bfloat16 if use_cuda
                          else torch.float32, channels_last=use_cuda)
            logits = model(x)
            c = logits.shape[1]
            loss = softmax_cross_entropy(
                logits.permute(0, 2, 3, 1).reshape(-1, c), y.reshape(-1))
        loss.backward()
        engine.finalize_backward()
        opt.step()
    feed.terminate()
    if ctx.is_chief:
        ctx.export_saved_model(model.cpu(), args.export_dir)
    torch.distributed.destroy_process_group()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--cluster_size", type=int, default=2)
    p.add_argument("--batch_size", type=int, default=8)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--num_gpus", type=int, default=1)
    p.add_argument("--export_dir", default="segmentation_export")
    args = p.parse_args()
    # executor working dirs differ from the driver's: path args
    # must be absolute (shared-filesystem semantics, as on a real
    # cluster)
    args.export_dir = os.path.abspath(args.export_dir)

    import numpy as np

    from tensorflowonspark_amd import TFCluster
    from tensorflowonspark_amd.local_context import LocalSparkContext
    sc = LocalSparkContext(num_executors=args.cluster_size)

    rng = np.random.default_rng(0)
    n = args.steps * args.batch_size * args.cluster_size
    rows = []
    for _ in range(n):
        img = rng.integers(0, 256, size=(128, 128, 3), dtype=np.uint8)
        mask = (img[:, :, 0] > 170).astype(np.int64) + \
               (img[:, :, 0] > 85).astype(np.int64)
        rows.append((img.reshape(-1), mask.reshape(-1)))

    cluster = TFCluster.run(sc, map_fun, args, args.cluster_size,
                            input_mode=TFCluster.InputMode.SPARK,
                            master_node="chief", num_gpus=args.num_gpus,
                            ring_slot_bytes=64 << 20)
    cluster.train(sc.parallelize(rows, args.cluster_size * 2), 1)
    cluster.shutdown(grace_secs=5)
    sc.stop()
    print("done")


if __name__ == "__main__":
    main()
