"""Streaming feed: DStream micro-batches -> cluster, stop via reservation STOP
(shape parity: reference streaming flow, ``mnist_spark_streaming.py`` +
``examples/utils/stop_streaming.py``)."""

import time

import pytest

from tensorflowonspark_amd import TFCluster, reservation
from tensorflowonspark_amd.local_context import (LocalSparkContext,
                                                 LocalStreamingContext)


def _stream_fn(args, ctx):
    feed = ctx.get_data_feed(train_mode=True)
    total = 0
    while not feed.should_stop():
        batch = feed.next_batch(50)
        if not batch:
            break
        total += sum(batch)
        # running total visible to the test
        with open("stream_sum.txt", "w") as f:
            f.write(str(total))


@pytest.mark.timeout(300)
def test_streaming_feed():
    sc = LocalSparkContext(num_executors=2)
    try:
        ssc = LocalStreamingContext(sc)
        cluster = TFCluster.run(sc, _stream_fn, {}, num_executors=2, num_ps=0,
                                master_node=None,
                                input_mode=TFCluster.InputMode.SPARK,
                                num_gpus=0, reservation_timeout=60)
        stream = ssc.queueStream()
        cluster.train(stream, feed_timeout=60)

        # push three micro-batches
        expected = 0
        for k in range(3):
            data = list(range(k * 100, k * 100 + 100))
            expected += sum(data)
            ssc.push(sc.parallelize(data, 2))
        time.sleep(3)  # let the worker drain

        # external stop: reservation STOP (as examples/utils/stop_streaming.py)
        client = reservation.Client(tuple(cluster.cluster_meta["server_addr"]))
        client.request_stop()
        client.close()

        cluster.shutdown(ssc=ssc, grace_secs=1)
        assert ssc._stopped.is_set()

        import glob
        import os
        total = sum(int(open(f).read()) for f in
                    glob.glob(os.path.join(sc._root, "executor_*",
                                           "stream_sum.txt")))
        assert total == expected
    finally:
        sc.stop()
