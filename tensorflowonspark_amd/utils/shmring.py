"""Shared-memory block ring: the InputMode.SPARK bulk-data path.

The reference pushed every RDD row through a ``multiprocessing.Manager`` proxy
queue — two picklings and two socket hops *per sample* (reference
``TFSparkNode.py:500-502`` producer, ``TFNode.py:279`` consumer); that is its
known feed bottleneck. Here the payload travels through POSIX shared memory in
*blocks* of many rows, and only tiny slot descriptors ride the manager queues
(which also preserves the reference's ``queue.join()`` completion semantics —
the feeder's ``join()`` returns only when the consumer has ``task_done()``-ed
every block descriptor).

Layout: one SharedMemory segment of ``slots * slot_bytes``. Producer:
``slot = ring.acquire()`` (from the ``free`` queue) → write payload →
``ring.commit(slot, n)`` (descriptor onto the ``data`` queue). Consumer:
``ring.take()`` → read payload → ``ring.release(slot)``.

On a GPU worker the consumer copies each block straight into a *pinned* staging
tensor and launches ``hipMemcpyAsync`` on a side stream (see
``TFNode.DataFeed.next_batch``), overlapping H2D with compute.
"""

import pickle
from multiprocessing import shared_memory


class _no_track:
    """Attach SharedMemory without resource-tracker registration.

    Attaching processes are not owners; CPython < 3.13 registers on attach
    anyway, which makes every attacher's tracker unlink the segment at exit
    (and later unregisters double-count, spamming KeyError warnings).
    Suppressing registration during attach avoids both.
    """

    def __enter__(self):
        from multiprocessing import resource_tracker
        self._rt = resource_tracker
        self._orig = resource_tracker.register
        resource_tracker.register = lambda *a, **k: None
        return self

    def __exit__(self, *exc):
        self._rt.register = self._orig


class BlockRing:
    """A ring of fixed-size shared-memory slots coordinated by two queues."""

    def __init__(self, name, slots, slot_bytes, data_queue, free_queue, create=False):
        self.slots = slots
        self.slot_bytes = slot_bytes
        self._data_q = data_queue
        self._free_q = free_queue
        if create:
            self.shm = shared_memory.SharedMemory(
                name=name, create=True, size=slots * slot_bytes)
            for i in range(slots):
                free_queue.put(i)
        else:
            with _no_track():
                self.shm = shared_memory.SharedMemory(name=name)
        self.name = self.shm.name
        self._owner = create

    # -- producer side -------------------------------------------------------

    def acquire(self, timeout=None):
        """Block until a free slot index is available."""
        return self._free_q.get(timeout=timeout) if timeout else self._free_q.get()

    def write(self, slot, payload):
        """Write raw bytes into a slot; returns length."""
        n = len(payload)
        if n > self.slot_bytes:
            raise ValueError(
                "payload {} B exceeds slot size {} B".format(n, self.slot_bytes))
        off = slot * self.slot_bytes
        self.shm.buf[off:off + n] = payload
        return n

    def commit(self, slot, nbytes, meta=None):
        """Publish a written slot to the consumer."""
        self._data_q.put((slot, nbytes, meta))
        # 'free' queue join-semantics: mark our own get() as processed so the
        # ring can be joined on the free queue as well if ever needed.
        self._free_q.task_done()

    def put_block(self, payload, meta=None, timeout=None):
        """acquire + write + commit in one call."""
        slot = self.acquire(timeout=timeout)
        n = self.write(slot, payload)
        self.commit(slot, n, meta)

    def put_rows(self, rows, meta=None):
        """Pickle a list of rows as one block (single pickling per block)."""
        self.put_block(pickle.dumps(rows, protocol=pickle.HIGHEST_PROTOCOL), meta=meta)

    def put_arrays(self, arrays, timeout=None):
        """Zero-pickle block of named numpy arrays (the fast tensor-feed path).

        Payload = raw array bytes back to back; the descriptor meta carries
        (name, dtype, shape, nbytes) per array. The consumer can read each
        array straight into a pinned staging tensor (``read_into``) for
        ``hipMemcpyAsync`` H2D.
        """
        slot = self.acquire(timeout=timeout)
        off = slot * self.slot_bytes
        meta = []
        pos = 0
        for name, arr in arrays.items():
            b = arr.tobytes() if not arr.flags["C_CONTIGUOUS"] else memoryview(arr).cast("B")
            n = len(b)
            if pos + n > self.slot_bytes:
                raise ValueError("array block exceeds slot size")
            self.shm.buf[off + pos:off + pos + n] = b
            meta.append((name, arr.dtype.str, arr.shape, n))
            pos += n
        self._data_q.put(("shm_arr", slot, pos, meta))
        self._free_q.task_done()

    def read_arrays(self, slot, meta, copy=True):
        """Materialize a put_arrays block as {name: np.ndarray}."""
        import numpy as np
        off = slot * self.slot_bytes
        out = {}
        pos = 0
        for name, dtype, shape, n in meta:
            view = np.frombuffer(self.shm.buf, dtype=np.dtype(dtype),
                                 count=int(np.prod(shape)) if shape else 1,
                                 offset=off + pos).reshape(shape)
            out[name] = view.copy() if copy else view
            pos += n
        return out

    # -- consumer side -------------------------------------------------------

    def take(self, timeout=None):
        """Get (slot, nbytes, meta) descriptor; None descriptor = end-of-feed."""
        desc = self._data_q.get(timeout=timeout) if timeout else self._data_q.get()
        return desc

    def read(self, slot, nbytes):
        """Return a *copy* of the slot payload as bytes."""
        off = slot * self.slot_bytes
        return bytes(self.shm.buf[off:off + nbytes])

    def read_into(self, slot, nbytes, dest):
        """Copy slot payload into a writable buffer (e.g. a pinned tensor)."""
        off = slot * self.slot_bytes
        dest[:nbytes] = self.shm.buf[off:off + nbytes]

    def release(self, slot):
        """Return the slot to the free pool and ack the descriptor."""
        self._free_q.put(slot)
        self._data_q.task_done()

    def take_rows(self, timeout=None):
        """Consume one pickled-rows block; returns (rows, meta) or (None, meta)."""
        desc = self.take(timeout=timeout)
        if desc is None:
            return None, None
        slot, nbytes, meta = desc
        rows = pickle.loads(self.read(slot, nbytes))
        self.release(slot)
        return rows, meta

    # -- lifecycle -----------------------------------------------------------

    def close(self):
        try:
            self.shm.close()
        except Exception:
            pass

    def unlink(self):
        if self._owner:
            try:
                self.shm.unlink()
            except Exception:
                pass
