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


def _cnn_train_fn(args, ctx):
    import torch

    from tensorflowonspark_amd.models import MNISTNet
    from tensorflowonspark_amd.ops.modules import BucketSGD, softmax_cross_entropy
    from tensorflowonspark_amd.parallel import DDPEngine
    ctx.init_process_group(backend="gloo")
    torch.manual_seed(0)
    model = MNISTNet()
    engine = DDPEngine(model, bucket_mb=2)
    opt = BucketSGD(engine, lr=0.05, momentum=0.9)
    feed = ctx.get_data_feed(train_mode=True)
    while True:
        batch = feed.next_batch(32) if not feed.should_stop() else []
        if not engine.all_ranks_ready(len(batch) > 0):
            break
        x = torch.tensor([r[0] for r in batch]).float().reshape(-1, 1, 28, 28) / 255
        y = torch.tensor([r[1] for r in batch])
        opt.zero_grad()
        loss = softmax_cross_entropy(model(x), y)
        loss.backward()
        engine.finalize_backward()
        opt.step()
    feed.terminate()
    if ctx.is_chief:
        ctx.export_saved_model(model.cpu(), args.export_dir)
    import torch.distributed as dist
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_transform_with_input_shapes(tmp_path):
    """Flat image arrays reshaped via the input_shapes param during transform
    (reference coerced shapes from the saved_model signature)."""
    import numpy as np
    sc = LocalSparkContext(num_executors=2)
    try:
        rng = np.random.default_rng(0)
        rows = [(rng.integers(0, 256, 784).tolist(), int(rng.integers(0, 10)))
                for _ in range(96)]
        df = sc.createDataFrame(rows, ["image", "label"])
        export_dir = str(tmp_path / "export")
        est = TFEstimator(_cnn_train_fn, {"export_dir": export_dir}) \
            .setClusterSize(2).setEpochs(1).setBatchSize(32) \
            .setInputMapping({"image": "x", "label": "y"})
        model = est.fit(df)

        tdf = sc.createDataFrame([(r[0],) for r in rows[:8]], ["image"])
        model.setInputMapping({"image": "x"}) \
             .setOutputMapping({"logits": "prediction"}) \
             .setInputShapes({"image": [1, 28, 28]})
        preds = model.transform(tdf).collect()
        assert len(preds) == 8
        assert len(preds[0][0]) == 10  # 10 logits per row
    finally:
        sc.stop()


def test_params_are_pyspark_params_when_available():
    """With pyspark installed, TFEstimator/TFModel must be real
    pyspark.ml.param.Params (and Pipeline stages); skipped otherwise."""
    import pytest
    pytest.importorskip("pyspark")
    from pyspark.ml import Estimator
    from pyspark.ml.param import Params

    from tensorflowonspark_amd.pipeline import TFEstimator

    est = TFEstimator(lambda a, c: None, {})
    assert isinstance(est, Params)
    assert isinstance(est, Estimator)
    est.setBatchSize(64)
    assert est.getBatchSize() == 64
    args = est.merge_args_params()
    assert args.batch_size == 64
