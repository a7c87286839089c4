"""DataFeed semantics against a real local TFManager
(shape parity: reference tests/test_TFNode.py:27-58)."""

import pickle

from tensorflowonspark_amd import TFManager, TFNode
from tensorflowonspark_amd.utils import shmring


def _mgr():
    return TFManager.start(b"k", ["input", "output", "error", "free"], "local")


def test_next_batch_inline_blocks():
    mgr = _mgr()
    try:
        q = mgr.get_queue("input")
        q.put(("rows", list(range(10))))
        q.put(None)
        feed = TFNode.DataFeed(mgr, train_mode=True)
        b1 = feed.next_batch(4)
        assert b1 == [0, 1, 2, 3]
        assert not feed.should_stop()
        b2 = feed.next_batch(4)
        assert b2 == [4, 5, 6, 7]
        b3 = feed.next_batch(4)
        assert b3 == [8, 9]          # short batch at end-of-feed
        assert feed.should_stop()
        q.join()                      # all items acked
    finally:
        mgr.shutdown()


def test_next_batch_shm_blocks():
    mgr = _mgr()
    try:
        ring = shmring.BlockRing("tfosr_feed_test", 4, 1 << 20,
                                 data_queue=mgr.get_queue("input"),
                                 free_queue=mgr.get_queue("free"), create=True)
        mgr.set("ring_name", ring.name)
        mgr.set("ring_slots", 4)
        mgr.set("ring_slot_bytes", 1 << 20)
        rows = [(i, i * 2.0) for i in range(100)]
        payload = pickle.dumps(rows)
        slot = ring.acquire()
        n = ring.write(slot, payload)
        mgr.get_queue("input").put(("shm", slot, n, len(rows)))
        mgr.get_queue("input").put(None)

        feed = TFNode.DataFeed(mgr, train_mode=True)
        batch = feed.next_batch(60)
        assert len(batch) == 60
        assert batch[0] == (0, 0.0)
        batch = feed.next_batch(60)
        assert len(batch) == 40
        assert feed.should_stop()
        ring.close()
        ring.unlink()
    finally:
        mgr.shutdown()


def test_end_partition_inference_semantics():
    mgr = _mgr()
    try:
        q = mgr.get_queue("input")
        q.put(("rows", [1, 2, 3]))
        q.put(("end_partition",))
        q.put(("rows", [4, 5]))
        q.put(None)
        feed = TFNode.DataFeed(mgr, train_mode=False)
        # inference mode: batch breaks at partition boundary
        b1 = feed.next_batch(10)
        assert b1 == [1, 2, 3]
        b2 = feed.next_batch(10)
        assert b2 == [4, 5]
        assert feed.should_stop()
    finally:
        mgr.shutdown()


def test_end_partition_train_mode_skipped():
    mgr = _mgr()
    try:
        q = mgr.get_queue("input")
        q.put(("rows", [1, 2]))
        q.put(("end_partition",))
        q.put(("rows", [3, 4]))
        q.put(None)
        feed = TFNode.DataFeed(mgr, train_mode=True)
        # train mode: markers are transparent
        assert feed.next_batch(4) == [1, 2, 3, 4]
    finally:
        mgr.shutdown()


def test_input_mapping_columnar():
    mgr = _mgr()
    try:
        q = mgr.get_queue("input")
        q.put(("rows", [(1, "a"), (2, "b")]))
        q.put(None)
        feed = TFNode.DataFeed(mgr, train_mode=True,
                               input_mapping={"col0": "x", "col1": "y"})
        batch = feed.next_batch(5)
        assert batch == {"x": [1, 2], "y": ["a", "b"]}
    finally:
        mgr.shutdown()


def test_batch_results_and_terminate():
    mgr = _mgr()
    try:
        feed = TFNode.DataFeed(mgr, train_mode=False)
        feed.batch_results(["r1", "r2"])
        out = mgr.get_queue("output")
        assert out.get() == "r1"
        out.task_done()
        assert out.get() == "r2"
        out.task_done()

        q = mgr.get_queue("input")
        q.put(("rows", [1]))
        q.put(("rows", [2]))
        feed.terminate()
        assert mgr.get("state") == "terminating"
        q.join()   # drained everything
    finally:
        mgr.shutdown()
