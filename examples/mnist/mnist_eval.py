#!/usr/bin/env python3
"""MNIST training with a dedicated evaluator node (reference pattern:
``examples/mnist/estimator/mnist_tf.py`` with ``eval_node=True`` — the
evaluator runs independently, polling ``model_dir`` for new checkpoints and
scoring them on held-out data while the workers train).

Run (local executors):
    python examples/mnist/mnist_eval.py --cluster_size 3 --eval_data eval.csv

Roles: 1 evaluator + (cluster_size-1) sync-SGD workers. The evaluator writes
``eval-<step>`` scalar events next to the checkpoints, so the chief-spawned
TensorBoard shows train AND eval curves.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))


def map_fun(args, ctx):
    import numpy as np
    import torch

    from tensorflowonspark_amd import TFNode
    from tensorflowonspark_amd.models import MNISTNet
    from tensorflowonspark_amd.ops.modules import (BucketSGD,
                                                   softmax_cross_entropy)
    from tensorflowonspark_amd.parallel import DDPEngine
    from tensorflowonspark_amd.utils import checkpoint as ckpt
    from tensorflowonspark_amd.utils.events import SummaryWriter

    if ctx.job_name == "evaluator":
        # --- evaluator: poll model_dir, score each new checkpoint ---
        data = np.loadtxt(args.eval_data, delimiter=",", dtype=np.float32)
        x = torch.as_tensor(data[:, 1:] / 255.0).reshape(-1, 1, 28, 28)
        y = torch.as_tensor(data[:, 0], dtype=torch.long)
        model = MNISTNet()
        writer = SummaryWriter(os.path.join(args.model_dir, "eval"))
        seen = -1
        idle = 0.0
        while idle < args.eval_timeout:
            path = ckpt.latest_checkpoint(args.model_dir)
            if path is None:
                time.sleep(1)
                idle += 1
                continue
            step, _ = ckpt.load_checkpoint(path, model)
            if step == seen:
                time.sleep(1)
                idle += 1
                continue
            seen, idle = step, 0.0
            model.eval()
            with torch.no_grad():
                logits = model(x)
                loss = float(torch.nn.functional.cross_entropy(logits, y))
                acc = float((logits.argmax(1) == y).float().mean())
            writer.add_scalars({"eval_loss": loss, "eval_acc": acc}, step)
            writer.flush()
            print("evaluator: step {} loss {:.4f} acc {:.3f}".format(
                step, loss, acc))
        writer.close()
        return

    # --- workers: sync-SGD training off the Spark feed ---
    ctx.init_process_group()
    device = ctx.device
    model = MNISTNet().to(device)
    engine = DDPEngine(model, bucket_mb=4)
    opt = BucketSGD(engine, lr=args.lr, momentum=0.9)
    feed = TFNode.DataFeed(ctx.mgr, train_mode=True)
    model.train()
    step = 0
    while True:
        batch = feed.next_batch(args.batch_size) if not feed.should_stop() else []
        if not engine.all_ranks_ready(len(batch) > 0):
            break
        arr = np.asarray(batch, dtype=np.float32)
        yb = torch.as_tensor(arr[:, 0], dtype=torch.long, device=device)
        xb = torch.as_tensor(arr[:, 1:] / 255.0, device=device).reshape(
            -1, 1, 28, 28)
        opt.zero_grad()
        loss = softmax_cross_entropy(model(xb), yb)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        step += 1
        if step % args.ckpt_every == 0 and ctx.is_chief:
            ckpt.save_checkpoint(args.model_dir, step, model)
    feed.terminate()
    if ctx.is_chief:
        ckpt.save_checkpoint(args.model_dir, step, model)
    torch.distributed.destroy_process_group()


def main():
    from tensorflowonspark_amd import TFCluster
    from tensorflowonspark_amd.local_context import LocalSparkContext

    p = argparse.ArgumentParser()
    p.add_argument("--cluster_size", type=int, default=3)
    p.add_argument("--batch_size", type=int, default=64)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--lr", type=float, default=0.05)
    p.add_argument("--data", default="mnist.csv")
    p.add_argument("--eval_data", default=None)
    p.add_argument("--model_dir", default="mnist_model")
    p.add_argument("--ckpt_every", type=int, default=20)
    p.add_argument("--eval_timeout", type=float, default=30)
    args = p.parse_args()
    args.model_dir = os.path.abspath(args.model_dir)

    if not os.path.exists(args.data):  # synthesize a tiny dataset
        import numpy as np
        d = os.path.dirname(os.path.abspath(args.data))
        if d:
            os.makedirs(d, exist_ok=True)
        rng = np.random.RandomState(0)
        rows = np.hstack([rng.randint(0, 10, (600, 1)),
                          rng.randint(0, 256, (600, 784))])
        np.savetxt(args.data, rows, fmt="%d", delimiter=",")
    if args.eval_data is None:
        args.eval_data = args.data
    args.eval_data = os.path.abspath(args.eval_data)

    sc = LocalSparkContext(num_executors=args.cluster_size)
    cluster = TFCluster.run(sc, map_fun, args, args.cluster_size, num_ps=0,
                            input_mode=TFCluster.InputMode.SPARK,
                            eval_node=True, num_gpus=0)
    import numpy as np
    data = np.loadtxt(args.data, delimiter=",", dtype=np.float32)
    rdd = sc.parallelize([tuple(r) for r in data], args.cluster_size - 1)
    cluster.train(rdd, num_epochs=args.epochs)
    cluster.shutdown(grace_secs=5)
    print("train done; eval events in", os.path.join(args.model_dir, "eval"))


if __name__ == "__main__":
    main()
