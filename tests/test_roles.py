"""Role coverage: evaluator node lifecycle + driver_ps_nodes async training."""

import os

import pytest

from tensorflowonspark_amd import TFCluster
from tensorflowonspark_amd.local_context import LocalSparkContext


def _eval_fn(args, ctx):
    if ctx.job_name == "evaluator":
        # evaluator runs independently (background process) and exits; the
        # executor's foreground loop then waits for the shutdown control msg
        with open("evaluator_ran.txt", "w") as f:
            f.write("yes task_index={}".format(ctx.task_index))
        return
    feed = ctx.get_data_feed(train_mode=True)
    while not feed.should_stop():
        if not feed.next_batch(10):
            break


@pytest.mark.timeout(300)
def test_evaluator_role():
    sc = LocalSparkContext(num_executors=3)
    try:
        cluster = TFCluster.run(sc, _eval_fn, {}, num_executors=3, num_ps=0,
                                master_node=None, eval_node=True,
                                input_mode=TFCluster.InputMode.SPARK,
                                num_gpus=0, reservation_timeout=60)
        roles = sorted(n["job_name"] for n in cluster.cluster_info)
        assert roles == ["evaluator", "worker", "worker"]
        cluster.train(sc.parallelize(range(100), 2), num_epochs=1)
        cluster.shutdown(grace_secs=1)
        import glob
        hits = glob.glob(os.path.join(sc._root, "executor_*", "evaluator_ran.txt"))
        assert len(hits) == 1
    finally:
        sc.stop()


def _dps_fn(args, ctx):
    import torch

    from tensorflowonspark_amd.parallel import DDPEngine
    from tensorflowonspark_amd.parallel.ps import AsyncSGD
    torch.manual_seed(ctx.executor_id)
    model = torch.nn.Linear(1, 1, bias=False)
    engine = DDPEngine(model, bucket_mb=1, broadcast_params=False)
    opt = AsyncSGD(engine, ctx.ps_client(), lr=0.05, momentum=0.0)
    feed = ctx.get_data_feed(train_mode=True)
    first = last = None
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
        last = loss.item()
        if first is None:
            first = last
    with open("dps_losses.txt", "w") as f:
        f.write("{} {}".format(first, last))


@pytest.mark.timeout(300)
def test_driver_ps_nodes():
    """ps runs as a driver thread; only num_executors-num_ps Spark executors
    are provisioned (the ps slot never lands on an executor)."""
    sc = LocalSparkContext(num_executors=2)
    try:
        cluster = TFCluster.run(sc, _dps_fn, {}, num_executors=3, num_ps=1,
                                master_node=None, driver_ps_nodes=True,
                                input_mode=TFCluster.InputMode.SPARK,
                                num_gpus=0, reservation_timeout=60)
        ps_nodes = [n for n in cluster.cluster_info if n["job_name"] == "ps"]
        assert len(ps_nodes) == 1 and ps_nodes[0].get("driver_ps")
        data = [((i % 10) / 10.0, 2.0 * (i % 10) / 10.0) for i in range(400)]
        cluster.train(sc.parallelize(data, 2), num_epochs=3)
        cluster.shutdown(grace_secs=1)
        import glob
        results = []
        for f in glob.glob(os.path.join(sc._root, "executor_*", "dps_losses.txt")):
            first, last = open(f).read().split()
            results.append((float(first), float(last)))
        assert len(results) == 2
        for first, last in results:
            assert last < first
    finally:
        sc.stop()


def _tf_mode_ps_fn(args, ctx):
    if ctx.job_name == "ps":
        ctx.run_parameter_server()
        return
    # TENSORFLOW-mode worker: no feed, just signal it ran
    with open("tfmode_worker.txt", "w") as f:
        f.write("ran")


@pytest.mark.timeout(300)
def test_tensorflow_mode_with_ps_shuts_down():
    """shutdown() must not wait forever on ps bootstrap tasks when workers
    read data directly (InputMode.TENSORFLOW completion signaling)."""
    sc = LocalSparkContext(num_executors=3)
    try:
        cluster = TFCluster.run(sc, _tf_mode_ps_fn, {}, num_executors=3,
                                num_ps=1, master_node=None,
                                input_mode=TFCluster.InputMode.TENSORFLOW,
                                num_gpus=0, reservation_timeout=60)
        cluster.shutdown(grace_secs=0)
        import glob
        hits = glob.glob(os.path.join(sc._root, "executor_*",
                                      "tfmode_worker.txt"))
        assert len(hits) == 2
    finally:
        sc.stop()
