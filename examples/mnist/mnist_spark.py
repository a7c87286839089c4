#!/usr/bin/env python3
"""MNIST CNN via InputMode.SPARK — the canonical end-to-end example
(parity: reference ``examples/mnist/keras/mnist_spark.py``).

CSV rows (label,784 pixels) are parallelized into an RDD and fed through the
shared-memory ring to GPU (or CPU) workers; workers train the reference's CNN
architecture under sync-SGD (RCCL all-reduce on GPU, gloo on CPU), checkpoint
per epoch to ``--model_dir`` and the chief exports to ``--export_dir``.

Run without Spark (local executor pool):
  python examples/mnist/mnist_spark.py --cluster_size 2 --epochs 2
With a real pyspark cluster, pass --use_pyspark and spark-submit this file.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def map_fun(args, ctx):
    import torch

    from tensorflowonspark_amd import TFNode
    from tensorflowonspark_amd.models import MNISTNet
    from tensorflowonspark_amd.ops.modules import BucketSGD, softmax_cross_entropy
    from tensorflowonspark_amd.parallel import DDPEngine
    from tensorflowonspark_amd.utils import checkpoint as ckpt
    from tensorflowonspark_amd.utils.metrics import StepTimer

    ctx.init_process_group()
    device = ctx.device
    model = MNISTNet().to(device)
    start_step, _ = ckpt.load_latest(args.model_dir, model)  # resume support
    engine = DDPEngine(model, bucket_mb=4)
    opt = BucketSGD(engine, lr=args.lr, momentum=0.9)
    feed = TFNode.DataFeed(ctx.mgr, train_mode=True)
    timer = StepTimer(args.batch_size, log_every=50)
    # chief writes TensorBoard scalar events so the TFCluster-spawned
    # TensorBoard (tensorboard_url) serves populated curves
    writer = None
    if ctx.is_chief and getattr(args, "model_dir", None):
        from tensorflowonspark_amd.utils.events import SummaryWriter
        writer = SummaryWriter(args.model_dir)
    model.train()

    step = start_step
    while True:
        batch = feed.next_batch(args.batch_size) if not feed.should_stop() else []
        # built-in uneven-partition guard: stop together once any rank runs dry
        if not engine.all_ranks_ready(len(batch) > 0):
            break
        import numpy as np
        arr = np.asarray(batch, dtype=np.float32)
        y = torch.as_tensor(arr[:, 0], dtype=torch.long, device=device)
        x = torch.as_tensor(arr[:, 1:] / 255.0, device=device).reshape(
            -1, 1, 28, 28)
        opt.zero_grad()
        loss = softmax_cross_entropy(model(x), y)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        timer.step()
        step += 1
        if writer is not None and step % 10 == 0:
            writer.add_scalars({"loss": float(loss),
                                "images_per_sec": timer.rate()}, step)
        if step % 200 == 0 and ctx.is_chief:
            ckpt.save_checkpoint(args.model_dir, step, model)

    # drain anything this rank didn't consume (reference mnist_spark.py:71 —
    # leftover queued blocks would otherwise trip the feeder's feed_timeout)
    feed.terminate()
    if writer is not None:
        writer.close()
    if ctx.is_chief:
        ckpt.save_checkpoint(args.model_dir, step, model)
        ctx.export_saved_model(model.cpu(), args.export_dir)
    torch.distributed.destroy_process_group()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--cluster_size", type=int, default=2)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--batch_size", type=int, default=64)
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--data", default="data/mnist/mnist.csv")
    p.add_argument("--model_dir", default="mnist_model")
    p.add_argument("--export_dir", default="mnist_export")
    p.add_argument("--num_gpus", type=int, default=1)
    p.add_argument("--use_pyspark", action="store_true")
    args = p.parse_args()
    # executor working dirs differ from the driver's: path args
    # must be absolute (shared-filesystem semantics, as on a real
    # cluster)
    args.model_dir = os.path.abspath(args.model_dir)
    args.export_dir = os.path.abspath(args.export_dir)

    from tensorflowonspark_amd import TFCluster
    if args.use_pyspark:
        from pyspark import SparkContext
        sc = SparkContext()
    else:
        from tensorflowonspark_amd.local_context import LocalSparkContext
        sc = LocalSparkContext(num_executors=args.cluster_size)

    if not os.path.exists(args.data):
        print("generating data (run mnist_data_setup.py for more control)")
        out_dir = os.path.dirname(args.data) or "."
        os.system("{} {} --output {} --format csv".format(
            sys.executable,
            os.path.join(os.path.dirname(os.path.abspath(__file__)),
                         "mnist_data_setup.py"), out_dir))
        generated = os.path.join(out_dir, "mnist.csv")
        if not os.path.exists(args.data) and os.path.exists(generated):
            args.data = generated  # setup always writes mnist.csv
    args.data = os.path.abspath(args.data)

    rows = []
    with open(args.data) as f:
        for line in f:
            rows.append([int(v) for v in line.strip().split(",")])
    rdd = sc.parallelize(rows, args.cluster_size * 2)

    cluster = TFCluster.run(sc, map_fun, args, args.cluster_size,
                            input_mode=TFCluster.InputMode.SPARK,
                            master_node="chief", num_gpus=args.num_gpus)
    cluster.train(rdd, num_epochs=args.epochs)
    cluster.shutdown(grace_secs=30)
    print("done; model_dir:", args.model_dir, "export:", args.export_dir)
    sc.stop()


if __name__ == "__main__":
    main()
