#!/usr/bin/env python3
"""Async-SGD ResNet with direct TFRecord reads — BASELINE config 4 shape
(ResNet-50 async-SGD InputMode.TENSORFLOW): ps-role nodes hold fp32 master
shards, workers read their TFRecord shard directly (no feeding job) and push
gradients without any cross-worker barrier.

CPU smoke:
  python examples/resnet/resnet_async.py --model resnet56_cifar \
      --cluster_size 3 --num_ps 1 --records 256 --num_gpus 0
On GPUs use --model resnet50 --cluster_size 5 --num_ps 1 (4 workers + 1 ps).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def map_fun(args, ctx):
    if ctx.job_name == "ps":
        ctx.run_parameter_server()
        return
    import numpy as np
    import torch

    from tensorflowonspark_amd.models import resnet50, resnet56_cifar
    from tensorflowonspark_amd.ops.modules import nhwc_pack, softmax_cross_entropy
    from tensorflowonspark_amd.parallel import DDPEngine
    from tensorflowonspark_amd.parallel.ps import AsyncSGD
    from tensorflowonspark_amd.utils.dataset import TFRecordDataset

    device = ctx.device
    use_cuda = device.type == "cuda"
    if args.model == "resnet50":
        model, shape = resnet50().to(device), (224, 224, 3)
    else:
        model, shape = resnet56_cifar().to(device), (32, 32, 3)
    if use_cuda:
        model = model.to(memory_format=torch.channels_last)
    model.train()
    engine = DDPEngine(model, broadcast_params=False)
    opt = AsyncSGD(engine, ctx.ps_client(), lr=args.lr, momentum=0.9)

    workers = len(ctx.cluster_spec.get("worker", [])) or 1
    ds = TFRecordDataset(args.data_dir, ctx.task_index, workers,
                         batch_size=args.batch_size, shuffle_buffer=256)
    amp = torch.autocast(device.type, dtype=torch.bfloat16, enabled=use_cuda)
    steps = 0
    for _epoch in range(args.epochs):
        for batch in ds:
            imgs = np.asarray([ex["image"][1] for ex in batch],
                              dtype=np.uint8).reshape((-1,) + shape)
            labs = np.asarray([ex["label"][1][0] for ex in batch],
                              dtype=np.int64)
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
            steps += 1
    print("worker {} done: {} async steps, loss {:.3f}".format(
        ctx.task_index, steps, float(loss)))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet56_cifar",
                   choices=["resnet50", "resnet56_cifar"])
    p.add_argument("--cluster_size", type=int, default=3)
    p.add_argument("--num_ps", type=int, default=1)
    p.add_argument("--batch_size", type=int, default=16)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--records", type=int, default=256)
    p.add_argument("--data_dir", default="")
    p.add_argument("--num_gpus", type=int, default=1)
    args = p.parse_args()

    import numpy as np

    from tensorflowonspark_amd import TFCluster, tfrecord
    from tensorflowonspark_amd.local_context import LocalSparkContext

    if not args.data_dir:
        # synthetic TFRecord shards (no network for real datasets)
        import tempfile
        args.data_dir = tempfile.mkdtemp(prefix="tfr_async_")
        shape = (224, 224, 3) if args.model == "resnet50" else (32, 32, 3)
        ncls = 1000 if args.model == "resnet50" else 10
        rng = np.random.default_rng(0)
        per_file = max(1, args.records // 4)
        i = 0
        for part in range(4):
            path = os.path.join(args.data_dir, "part-r-{:05d}".format(part))
            with tfrecord.TFRecordWriter(path) as w:
                for _ in range(per_file):
                    if i >= args.records:
                        break
                    w.write(tfrecord.encode_example({
                        "image": rng.integers(0, 256, np.prod(shape))
                                    .astype(np.int64).tolist(),
                        "label": int(rng.integers(0, ncls))}))
                    i += 1
        print("wrote synthetic TFRecords to", args.data_dir)

    sc = LocalSparkContext(num_executors=args.cluster_size)
    cluster = TFCluster.run(sc, map_fun, args, args.cluster_size,
                            num_ps=args.num_ps, master_node=None,
                            input_mode=TFCluster.InputMode.TENSORFLOW,
                            num_gpus=args.num_gpus)
    cluster.shutdown(grace_secs=2)
    sc.stop()
    print("async run complete")


if __name__ == "__main__":
    main()
