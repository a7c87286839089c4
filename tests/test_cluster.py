"""End-to-end cluster lifecycle on the Spark-free local executor pool
(shape parity: reference tests/test_TFCluster.py — independent instances,
InputMode.SPARK round trip, error propagation during and after feeding)."""

import pytest

from tensorflowonspark_amd import TFCluster
from tensorflowonspark_amd.local_context import LocalSparkContext


@pytest.fixture()
def sc():
    ctx = LocalSparkContext(num_executors=2)
    yield ctx
    ctx.stop()


def _square_fn(args, ctx):
    """Consume ints from the feed, square on the 'GPU worker', return results."""
    feed = ctx.get_data_feed(train_mode=False)
    while not feed.should_stop():
        batch = feed.next_batch(10)
        if not batch:
            break
        feed.batch_results([x * x for x in batch])


def test_inference_roundtrip(sc):
    cluster = TFCluster.run(sc, _square_fn, {}, num_executors=2, num_ps=0,
                            master_node=None, input_mode=TFCluster.InputMode.SPARK,
                            num_gpus=0, reservation_timeout=60)
    data = list(range(1000))
    rdd = sc.parallelize(data, 4)
    out = cluster.inference(rdd).collect()
    assert sorted(out) == sorted(x * x for x in data)
    cluster.shutdown(grace_secs=0)


def _sum_train_fn(args, ctx):
    feed = ctx.get_data_feed(train_mode=True)
    total = 0
    while not feed.should_stop():
        batch = feed.next_batch(100)
        total += sum(batch)
    # write result where the test can see it (executor cwd)
    with open("sum_result.txt", "w") as f:
        f.write(str(total))


def test_train_feed(sc):
    cluster = TFCluster.run(sc, _sum_train_fn, {}, num_executors=2, num_ps=0,
                            master_node=None, input_mode=TFCluster.InputMode.SPARK,
                            num_gpus=0, reservation_timeout=60)
    data = list(range(100))
    rdd = sc.parallelize(data, 2)
    cluster.train(rdd, num_epochs=2)
    cluster.shutdown(grace_secs=1)
    # both executors together consumed 2 epochs of the data
    import glob
    import os
    total = 0
    for f in glob.glob(os.path.join(sc._root, "executor_*", "sum_result.txt")):
        with open(f) as fh:
            total += int(fh.read())
    assert total == 2 * sum(data)


def _failing_fn(args, ctx):
    feed = ctx.get_data_feed(train_mode=True)
    feed.next_batch(1)
    raise RuntimeError("injected worker failure")


def test_error_during_feeding(sc):
    cluster = TFCluster.run(sc, _failing_fn, {}, num_executors=2, num_ps=0,
                            master_node=None, input_mode=TFCluster.InputMode.SPARK,
                            num_gpus=0, reservation_timeout=60)
    rdd = sc.parallelize(range(1000), 2)
    with pytest.raises(Exception, match="injected worker failure"):
        cluster.train(rdd, num_epochs=5, feed_timeout=30)
        cluster.shutdown(grace_secs=0)


def _late_failing_fn(args, ctx):
    feed = ctx.get_data_feed(train_mode=True)
    while not feed.should_stop():
        if not feed.next_batch(100):
            break
    raise RuntimeError("late failure after feeding")


def test_error_after_feeding_caught_by_shutdown(sc):
    cluster = TFCluster.run(sc, _late_failing_fn, {}, num_executors=2, num_ps=0,
                            master_node=None, input_mode=TFCluster.InputMode.SPARK,
                            num_gpus=0, reservation_timeout=60)
    rdd = sc.parallelize(range(100), 2)
    cluster.train(rdd, num_epochs=1)
    with pytest.raises(Exception, match="late failure"):
        cluster.shutdown(grace_secs=3)


def _guarded_train_fn(args, ctx):
    from tensorflowonspark_amd.parallel import DDPEngine
    import torch
    ctx.init_process_group(backend="gloo")
    model = torch.nn.Linear(2, 1)
    engine = DDPEngine(model, bucket_mb=1)
    steps = 0
    feed = ctx.get_data_feed(train_mode=True)
    while True:
        batch = feed.next_batch(40) if not feed.should_stop() else []
        if not engine.all_ranks_ready(len(batch) > 0):
            break
        steps += 1
    feed.terminate()  # drain unconsumed blocks so the feeder's join returns
    with open("guard_steps.txt", "w") as f:
        f.write(str(steps))
    torch.distributed.destroy_process_group()


def test_uneven_feed_through_cluster(sc):
    """3 partitions over 2 workers: one rank gets ~2x the data; the guard must
    stop both together and terminate() must unblock the feeder."""
    cluster = TFCluster.run(sc, _guarded_train_fn, {}, num_executors=2,
                            num_ps=0, master_node="chief",
                            input_mode=TFCluster.InputMode.SPARK,
                            num_gpus=0, reservation_timeout=60)
    rdd = sc.parallelize(range(600), 3)
    cluster.train(rdd, num_epochs=1, feed_timeout=60)
    cluster.shutdown(grace_secs=1)
    import glob
    import os
    import time
    # workers finish terminate()'s drain (5 s empty-queue timeout) after
    # shutdown returns; poll for their results
    deadline = time.time() + 30
    steps = []
    while time.time() < deadline:
        steps = [int(open(f).read()) for f in
                 glob.glob(os.path.join(sc._root, "executor_*",
                                        "guard_steps.txt"))]
        if len(steps) == 2:
            break
        time.sleep(0.5)
    assert len(steps) == 2
    assert steps[0] == steps[1], steps  # ranks stopped together


def _multi_queue_fn(args, ctx):
    """Consume from a custom queue name (reference supports arbitrary queues
    via TFCluster.run(queues=[...]) + qname on train/inference)."""
    feed = ctx.get_data_feed(train_mode=False, qname_in="alt_in",
                             qname_out="alt_out")
    while not feed.should_stop():
        batch = feed.next_batch(10)
        if not batch:
            break
        feed.batch_results([x + 1000 for x in batch])


def test_custom_queue_names(sc):
    cluster = TFCluster.run(sc, _multi_queue_fn, {}, num_executors=2, num_ps=0,
                            master_node=None,
                            input_mode=TFCluster.InputMode.SPARK, num_gpus=0,
                            queues=["alt_in", "alt_out", "error"],
                            reservation_timeout=60)
    rdd = sc.parallelize(range(100), 2)
    out = cluster.inference(rdd, qname="alt_in", qname_out="alt_out").collect()
    assert sorted(out) == [x + 1000 for x in range(100)]
    cluster.shutdown(grace_secs=0)


def _port_check_fn(args, ctx):
    """release_port=False: the reserved port stays bound until the user frees
    it (reference tests/test_TFCluster.py:93-121 semantics)."""
    import socket
    held = ctx.tmp_socket is not None
    bound_before = False
    if held:
        try:
            probe = socket.socket()
            probe.bind(("127.0.0.1", ctx._port))
            probe.close()
        except OSError:
            bound_before = True   # port is occupied by our reservation
    ctx.release_port()
    released = ctx.tmp_socket is None
    feed = ctx.get_data_feed(train_mode=False)
    while not feed.should_stop():
        batch = feed.next_batch(10)
        if not batch:
            break
        feed.batch_results([(held, bound_before, released)] * len(batch))


def test_release_port_deferred(sc):
    cluster = TFCluster.run(sc, _port_check_fn, {}, num_executors=2, num_ps=0,
                            master_node=None, release_port=False,
                            input_mode=TFCluster.InputMode.SPARK,
                            num_gpus=0, reservation_timeout=60)
    out = cluster.inference(sc.parallelize(range(4), 2)).collect()
    assert all(held and bound and released for held, bound, released in out), out
    cluster.shutdown(grace_secs=0)
