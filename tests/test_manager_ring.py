"""TFManager + shared-memory block-ring tests."""

import multiprocessing
import pickle

from tensorflowonspark_amd import TFManager
from tensorflowonspark_amd.utils import shmring


def _make_mgr(queues=("input", "output", "error", "free")):
    return TFManager.start(b"testkey", list(queues), "local")


def test_manager_kv_plain_values():
    mgr = _make_mgr()
    try:
        mgr.set("state", "running")
        assert mgr.get("state") == "running"          # plain str, not a proxy repr
        mgr.set("ring_slots", 8)
        assert mgr.get("ring_slots") + 1 == 9          # plain int arithmetic
        assert mgr.get("missing") is None
    finally:
        mgr.shutdown()


def test_manager_queues_and_connect():
    mgr = _make_mgr()
    try:
        q = mgr.get_queue("input")
        q.put(("rows", [1, 2, 3]))
        # connect from the same process via address
        m2 = TFManager.connect(mgr.address, b"testkey")
        item = m2.get_queue("input").get()
        assert item == ("rows", [1, 2, 3])
        m2.get_queue("input").task_done()
        q.join()  # returns because consumer acked
    finally:
        mgr.shutdown()


def test_manager_handle_pickles():
    mgr = _make_mgr()
    try:
        mgr.set("k", "v")
        h2 = pickle.loads(pickle.dumps(mgr))
        assert h2.get("k") == "v"
    finally:
        mgr.shutdown()


def _producer(address, authkey, ring_name, slots, slot_bytes, nblocks):
    mgr = TFManager.connect(address, authkey)
    ring = shmring.BlockRing(ring_name, slots, slot_bytes,
                             data_queue=mgr.get_queue("input"),
                             free_queue=mgr.get_queue("free"), create=False)
    for b in range(nblocks):
        rows = [(b, i, float(i) * b) for i in range(100)]
        ring.put_rows(rows, meta=b)
    ring.close()


def test_blockring_cross_process():
    mgr = _make_mgr()
    try:
        ring = shmring.BlockRing("tfosr_test_ring", 4, 1 << 20,
                                 data_queue=mgr.get_queue("input"),
                                 free_queue=mgr.get_queue("free"), create=True)
        nblocks = 10
        p = multiprocessing.Process(
            target=_producer,
            args=(mgr.address, b"testkey", ring.name, 4, 1 << 20, nblocks))
        p.start()
        total = 0
        for _ in range(nblocks):
            rows, meta = ring.take_rows(timeout=30)
            assert rows is not None
            assert len(rows) == 100
            total += len(rows)
        p.join(timeout=10)
        assert p.exitcode == 0
        assert total == nblocks * 100
        ring.close()
        ring.unlink()
    finally:
        mgr.shutdown()
