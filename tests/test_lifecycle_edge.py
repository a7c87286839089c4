"""Lifecycle corner cases the reference encodes as years of bug fixes
(survey §7 hard part 4): retry poisoning, duplicate registration, feed
timeout."""

import pytest

from tensorflowonspark_amd import TFCluster, TFManager, TFSparkNode, reservation


def test_stale_manager_poisons_retried_bootstrap():
    """A live TFManager with the same cluster_id must make a retried bootstrap
    task raise, pushing Spark to another executor
    (reference TFSparkNode.py:258-265)."""
    mgr = TFManager.start(b"k", ["input", "error"], "local")
    try:
        TFSparkNode.TFSparkNode.mgr = mgr
        TFSparkNode.TFSparkNode.owned_mgr = mgr
        TFSparkNode.TFSparkNode.cluster_id = 42
        mgr.set("state", "running")
        fn = TFSparkNode.run(lambda a, c: None, {}, {
            "id": 42, "cluster_template": {"worker": [0]},
            "default_fs": "file://", "working_dir": ".",
            "server_addr": ["127.0.0.1", 1], "num_gpus": 0})
        with pytest.raises(Exception, match="already running"):
            fn(iter([0]))
    finally:
        TFSparkNode.TFSparkNode.mgr = None
        TFSparkNode.TFSparkNode.owned_mgr = None
        TFSparkNode.TFSparkNode.cluster_id = None
        mgr.shutdown()


def test_duplicate_reservation_detected_at_run():
    """Two nodes registering the same (host, executor_id) fail cluster startup
    (reference TFCluster.py:357-372)."""
    server = reservation.Server(2)
    addr = server.start()
    c = reservation.Client(addr)
    meta = {"executor_id": 0, "host": "1.2.3.4", "job_name": "worker",
            "task_index": 0, "port": 1, "addr": ["1.2.3.4", 2], "authkey": ""}
    c.register(meta)
    c.register(dict(meta))  # same identity again (retried task that slipped by)
    roster = server.await_reservations(timeout=10)
    seen = set()
    dup = False
    for node in roster:
        key = (node["host"], node["executor_id"])
        dup = dup or key in seen
        seen.add(key)
    assert dup, "duplicate should be detectable from the roster"
    c.close()
    server.stop()


def test_feed_timeout_raises(tmp_path):
    """A consumer that never drains the queue trips feed_timeout
    (reference TFSparkNode.py:507-515)."""
    import os

    from tensorflowonspark_amd import util
    mgr = TFManager.start(b"k2", ["input", "output", "error", "free"], "local")
    try:
        mgr.set("state", "running")
        cwd = os.getcwd()
        os.chdir(tmp_path)
        try:
            util.write_executor_id(0)
            cluster_info = [{"host": "127.0.0.1", "executor_id": 0,
                             "addr": list(mgr.address), "authkey": b"k2".hex(),
                             "job_name": "worker", "task_index": 0, "port": 1}]
            os.environ["TFOS_FORCE_LOOPBACK"] = "1"
            fn = TFSparkNode.train(cluster_info, {"server_addr": ["127.0.0.1", 1],
                                                  "block_rows": 8},
                                   feed_timeout=3)
            with pytest.raises(Exception, match="timed out"):
                fn(iter(range(20)))
        finally:
            os.chdir(cwd)
            os.environ.pop("TFOS_FORCE_LOOPBACK", None)
    finally:
        mgr.shutdown()


def test_error_queue_peek_and_requeue():
    """Shutdown must re-put the error so a Spark retry still observes it
    (reference TFSparkNode.py:644-650)."""
    import os
    mgr = TFManager.start(b"k3", ["input", "output", "error", "free"], "local")
    try:
        mgr.set("state", "running")
        mgr.get_queue("error").put("trapped traceback")
        import tempfile
        d = tempfile.mkdtemp()
        cwd = os.getcwd()
        os.chdir(d)
        try:
            from tensorflowonspark_amd import util
            util.write_executor_id(7)
            os.environ["TFOS_FORCE_LOOPBACK"] = "1"
            cluster_info = [{"host": "127.0.0.1", "executor_id": 7,
                             "addr": list(mgr.address), "authkey": b"k3".hex(),
                             "job_name": "worker", "task_index": 0, "port": 1,
                             "tb_pid": 0}]
            fn = TFSparkNode.shutdown(cluster_info, ["input", "output", "error"])
            with pytest.raises(Exception, match="trapped traceback"):
                fn(iter([7]))
            # error is still there for the next retry
            q = mgr.get_queue("error")
            assert q.get(timeout=5) == "trapped traceback"
            q.task_done()
        finally:
            os.chdir(cwd)
            os.environ.pop("TFOS_FORCE_LOOPBACK", None)
    finally:
        mgr.shutdown()
