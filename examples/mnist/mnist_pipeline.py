#!/usr/bin/env python3
"""ML-pipeline MNIST: TFEstimator.fit -> TFModel.transform
(parity: reference ``examples/mnist/keras/mnist_pipeline.py``)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def train_fn(args, ctx):
    import torch

    from tensorflowonspark_amd.models import MNISTMLP
    from tensorflowonspark_amd.ops.modules import BucketSGD, softmax_cross_entropy
    from tensorflowonspark_amd.parallel import DDPEngine

    ctx.init_process_group()
    model = MNISTMLP().to(ctx.device)
    engine = DDPEngine(model, bucket_mb=4)
    opt = BucketSGD(engine, lr=0.05, momentum=0.9)
    feed = ctx.get_data_feed(train_mode=True)
    while True:
        batch = feed.next_batch(args.batch_size) if not feed.should_stop() else []
        if not engine.all_ranks_ready(len(batch) > 0):
            break
        import numpy as np
        img = np.asarray([r[0] for r in batch], dtype=np.float32) / 255.0
        lab = np.asarray([r[1] for r in batch], dtype=np.int64)
        x = torch.as_tensor(img, device=ctx.device)
        y = torch.as_tensor(lab, device=ctx.device)
        opt.zero_grad()
        loss = softmax_cross_entropy(model(x), y)
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
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--export_dir", default="mnist_pipeline_export")
    p.add_argument("--num", type=int, default=2000)
    args = p.parse_args()
    # executor working dirs differ from the driver's: path args
    # must be absolute (shared-filesystem semantics, as on a real
    # cluster)
    args.export_dir = os.path.abspath(args.export_dir)

    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "mnist_data_setup",
        os.path.join(os.path.dirname(os.path.abspath(__file__)),
                     "mnist_data_setup.py"))
    mds = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mds)
    synthetic_mnist = mds.synthetic_mnist
    from tensorflowonspark_amd.local_context import LocalSparkContext
    from tensorflowonspark_amd.pipeline import TFEstimator

    sc = LocalSparkContext(num_executors=args.cluster_size)
    images, labels = synthetic_mnist(args.num)
    rows = [(img.reshape(-1).tolist(), int(lab))
            for img, lab in zip(images, labels)]
    df = sc.createDataFrame(rows, ["image", "label"])

    est = TFEstimator(train_fn, {"export_dir": args.export_dir}) \
        .setClusterSize(args.cluster_size).setEpochs(args.epochs) \
        .setBatchSize(64).setInputMapping({"image": "x", "label": "y"})
    model = est.fit(df)

    test_rows = [(img.reshape(-1).tolist(),) for img in images[:64]]
    tdf = sc.createDataFrame(test_rows, ["image"])
    model.setInputMapping({"image": "x"}) \
         .setOutputMapping({"logits": "prediction"})
    preds = model.transform(tdf).collect()
    import numpy as np
    acc = np.mean([int(np.argmax(p[0])) == labels[i]
                   for i, (p,) in enumerate(zip(preds))])
    print("train-set head accuracy:", acc)
    sc.stop()


if __name__ == "__main__":
    main()
