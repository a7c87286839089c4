"""TFEstimator -> fit -> export -> TFModel.transform regression
(shape parity: reference tests/test_pipeline.py:89-172 — synthetic linear data
y = x . [3.14, 1.618], 1-dense model trained distributed, then transform)."""

import os

import pytest

from tensorflowonspark_amd import TFCluster
from tensorflowonspark_amd.local_context import LocalSparkContext
from tensorflowonspark_amd.pipeline import Namespace, TFEstimator, TFParams


def test_namespace():
    n = Namespace({"a": 1, "b": "x"})
    assert n.a == 1 and n.b == "x"
    assert n.undefined is None
    argv = Namespace(["--foo", "1"])
    assert argv.ARGV == ["--foo", "1"]


def test_params_merge():
    p = TFParams({"batch_size": 10, "custom": "keep"})
    p.setBatchSize(64).setClusterSize(2).setEpochs(3)
    args = p.merge_args_params()
    assert args.batch_size == 64       # param overlays arg
    assert args.cluster_size == 2
    assert args.epochs == 3
    assert args.custom == "keep"       # untouched arg survives


WEIGHTS = [3.14, 1.618]


def _train_fn(args, ctx):
    import torch

    from tensorflowonspark_amd.ops.modules import BucketSGD
    from tensorflowonspark_amd.parallel import DDPEngine

    ctx.init_process_group(backend="gloo")
    torch.manual_seed(0)
    model = torch.nn.Linear(2, 1, bias=False)
    engine = DDPEngine(model, bucket_mb=1)
    opt = BucketSGD(engine, lr=0.2, momentum=0.0)
    feed = ctx.get_data_feed(train_mode=True)
    while True:
        batch = feed.next_batch(args.batch_size) if not feed.should_stop() else []
        if not engine.all_ranks_ready(len(batch) > 0):
            break
        x = torch.tensor([r[0] for r in batch], dtype=torch.float32)
        y = torch.tensor([[r[1]] for r in batch], dtype=torch.float32)
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        engine.finalize_backward()
        opt.step()
    if ctx.is_chief:
        ctx.export_saved_model(model, args.export_dir)
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_estimator_fit_transform(tmp_path):
    sc = LocalSparkContext(num_executors=2)
    try:
        import numpy as np
        rng = np.random.default_rng(0)
        X = rng.normal(size=(400, 2))
        Y = X @ np.array(WEIGHTS)
        rows = [([float(a), float(b)], float(y)) for (a, b), y in zip(X, Y)]
        df = sc.createDataFrame(rows, ["features", "label"])

        export_dir = str(tmp_path / "export")
        est = TFEstimator(_train_fn, {"export_dir": export_dir}) \
            .setClusterSize(2).setEpochs(4).setBatchSize(32) \
            .setInputMapping({"features": "x", "label": "y"})
        model = est.fit(df)
        assert os.path.exists(os.path.join(export_dir, "model.pt"))

        test_rows = [([1.0, 0.0],), ([0.0, 1.0],), ([2.0, 3.0],)]
        tdf = sc.createDataFrame(test_rows, ["features"])
        model.setInputMapping({"features": "x"}) \
             .setOutputMapping({"output": "prediction"})
        preds = model.transform(tdf).collect()
        expect = [3.14, 1.618, 2 * 3.14 + 3 * 1.618]
        for (p,), e in zip(preds, expect):
            assert p[0] == pytest.approx(e, abs=0.05), (p, e)
    finally:
        sc.stop()
