"""Async parameter-server mode: unit test + full-cluster async training."""

import threading

import numpy as np
import pytest

from tensorflowonspark_amd.parallel import ps as ps_mod


def test_ps_server_roundtrip():
    server = ps_mod.ParameterServer(port=0)
    t = threading.Thread(target=server.serve_forever, daemon=True)
    t.start()
    client = ps_mod.PSClient(["127.0.0.1:{}".format(server.port)])

    p0 = np.arange(10, dtype=np.float32)
    client.init_bucket(0, p0)
    out = np.empty(10, dtype=np.float32)
    client.pull(0, out)
    assert np.allclose(out, p0)

    grad = np.ones(10, dtype=np.float32)
    client.push_pull(0, grad, out, lr=0.5, momentum=0.0, weight_decay=0.0)
    assert np.allclose(out, p0 - 0.5)
    # second push accumulates momentum: m = 0.9*1 + 1 = 1.9
    client.push_pull(0, grad, out, lr=0.5, momentum=0.9, weight_decay=0.0)
    assert np.allclose(out, p0 - 0.5 - 0.5 * 1.9)
    client.stop_all()
    client.close()
    t.join(timeout=5)
    assert not t.is_alive()


def _ps_fn(args, ctx):
    if ctx.job_name == "ps":
        ctx.run_parameter_server()
        return
    # worker: async SGD on y = 2x linear data via the feed
    import torch

    from tensorflowonspark_amd.ops.modules import softmax_cross_entropy  # noqa
    from tensorflowonspark_amd.parallel import DDPEngine
    from tensorflowonspark_amd.parallel.ps import AsyncSGD

    torch.manual_seed(ctx.executor_id)
    model = torch.nn.Linear(1, 1, bias=False)
    engine = DDPEngine(model, bucket_mb=1, broadcast_params=False)
    opt = AsyncSGD(engine, ctx.ps_client(), lr=0.05, momentum=0.0)
    feed = ctx.get_data_feed(train_mode=True)
    losses = []
    while not feed.should_stop():
        batch = feed.next_batch(16)
        if not batch:
            break
        x = torch.tensor([[r[0]] for r in batch])
        y = torch.tensor([[r[1]] for r in batch])
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        losses.append(loss.item())
    with open("ps_losses.txt", "w") as f:
        f.write("{} {}".format(losses[0], losses[-1]))


@pytest.mark.timeout(300)
def test_async_ps_training():
    from tensorflowonspark_amd import TFCluster
    from tensorflowonspark_amd.local_context import LocalSparkContext
    sc = LocalSparkContext(num_executors=3)
    try:
        cluster = TFCluster.run(sc, _ps_fn, {}, num_executors=3, num_ps=1,
                                master_node=None,
                                input_mode=TFCluster.InputMode.SPARK,
                                num_gpus=0, reservation_timeout=60)
        data = [((i % 10) / 10.0, 2.0 * (i % 10) / 10.0) for i in range(400)]
        rdd = sc.parallelize(data, 2)
        cluster.train(rdd, num_epochs=3)
        cluster.shutdown(grace_secs=1)
        import glob
        import os
        results = []
        for f in glob.glob(os.path.join(sc._root, "executor_*", "ps_losses.txt")):
            first, last = open(f).read().split()
            results.append((float(first), float(last)))
        assert len(results) == 2, "both workers should have trained"
        for first, last in results:
            assert last < first, "async training did not reduce loss"
    finally:
        sc.stop()


def test_async_sgd_overlap_and_bf16_wire():
    """push/pull round trips start from the grad-ready
    hooks during backward (not a serial post-step loop), and the bf16 wire
    format round-trips through the server."""
    import threading

    import numpy as np
    import torch

    from tensorflowonspark_amd.parallel import DDPEngine
    from tensorflowonspark_amd.parallel.ps import (AsyncSGD, ParameterServer,
                                                   PSClient, _bf16_to_f32,
                                                   _f32_to_bf16)

    # wire converters round trip within bf16 precision
    a = np.linspace(-3, 3, 64, dtype=np.float32)
    b = _bf16_to_f32(_f32_to_bf16(a))
    assert np.abs(a - b).max() < 0.02

    srv = ParameterServer()
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    client = PSClient(["127.0.0.1:{}".format(srv.port)])
    model = torch.nn.Linear(1, 1, bias=False)
    eng = DDPEngine(model, bucket_mb=1, broadcast_params=False)
    opt = AsyncSGD(eng, client, lr=0.2, momentum=0.0, wire="bf16")
    assert eng.bucket_ready_cb is not None  # overlap installed

    for _ in range(150):
        x = torch.rand(32, 1)
        y = 2 * x
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        # round trips already in flight before step() (overlap)
        in_flight = len(opt._threads)
        eng.finalize_backward()
        opt.step()
        assert in_flight >= 1
    w = float(model.weight.detach().flatten()[0])
    assert abs(w - 2.0) < 0.1, w
    client.stop_all()
    srv.stop()
