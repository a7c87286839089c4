"""User-facing helpers inside ``map_fun`` (parity: reference ``TFNode.py``).

``DataFeed`` is the InputMode.SPARK consumer API with the reference's semantics
(``next_batch``/``batch_results``/``should_stop``/``terminate``, ``None`` =
end-of-feed, ``EndPartition`` = partition boundary, reference ``TFNode.py:234-343``)
— but consuming shared-memory *blocks* instead of per-row queue items, and able
to hand back pinned torch tensors ready for ``hipMemcpyAsync`` H2D on a side
stream.

``start_cluster_server`` (TF1 gRPC server) has no MI355X analog: collectives are
RCCL via ``torch.distributed`` — call ``ctx.init_process_group()`` instead.
"""

import getpass
import logging
import os
from collections import deque

logger = logging.getLogger(__name__)

_HADOOP_SCHEMES = ["adl://", "file://", "gs://", "hdfs://", "oss://", "s3://",
                   "s3a://", "s3n://", "swift://", "viewfs://", "wasb://", "abfs://"]


def hdfs_path(ctx, path):
    """Make ``path`` absolute relative to the cluster's default filesystem.

    Scheme table parity with reference ``TFNode.py:32-67``: known schemes pass
    through; absolute paths are prefixed with defaultFS; relative paths resolve
    under ``hdfs://…/user/<user>/`` or ``file://<cwd>/``.
    """
    for scheme in _HADOOP_SCHEMES:
        if path.startswith(scheme):
            return path
    fs = getattr(ctx, "defaultFS", "file://")
    if path.startswith("/"):
        return fs + path
    if fs.startswith("hdfs://") or fs.startswith("viewfs://"):
        return "{}/user/{}/{}".format(fs, getpass.getuser(), path)
    if fs.startswith("file://"):
        return "{}/{}/{}".format(fs, getattr(ctx, "working_dir", os.getcwd())[1:], path)
    logger.warning("unknown scheme '%s' for path '%s'", fs, path)
    return "{}/{}".format(fs, path)


def start_cluster_server(ctx, num_gpus=1, rdma=False):
    """Unsupported: the MI355X data plane is RCCL via torch.distributed."""
    raise NotImplementedError(
        "start_cluster_server is a TF1/gRPC concept; use ctx.init_process_group() "
        "— collectives run over RCCL/xGMI via torch.distributed.")


def export_saved_model(model, export_dir, is_chief=True, require_script=False):
    """Chief-only model export (layout parity: versioned export_dir).

    Saves a TorchScript export when possible, else the state_dict, under
    ``export_dir``. Non-chief ranks write nothing (reference ``compat.py:10-17``
    had non-chief write to a dummy path; skipping is the cleaner equivalent).

    A state_dict fallback cannot be consumed by ``TFModel.transform`` (it
    needs TorchScript); pass ``require_script=True`` to fail *here* instead of
    at transform time.
    """
    if not is_chief:
        return None
    import copy

    import torch
    os.makedirs(export_dir, exist_ok=True)
    try:
        # script a deepcopy: torch.jit.script destructively swaps children
        # via __prepare_scriptable__ (the fused/MFMA modules provide plain-op
        # clones there) — the live training model must stay untouched. The
        # export is a serving artifact, so it is scripted in eval mode.
        scripted = torch.jit.script(copy.deepcopy(model).eval())
        path = os.path.join(export_dir, "model.pt")
        scripted.save(path)
    except Exception as e:
        if require_script:
            raise RuntimeError(
                "torch.jit.script failed for this model and the export is "
                "required to be TorchScript (TFModel.transform consumes it): "
                "{}".format(e)) from e
        path = os.path.join(export_dir, "state_dict.pt")
        torch.save(model.state_dict(), path)
        logger.warning(
            "torch.jit.script failed (%s); exported state_dict only to %s — "
            "TFModel.transform will NOT be able to load this export", e, path)
    finally:
        # release the originals the __prepare_scriptable__ hooks kept alive
        # (guards torch's id()-memoized prepare recursion against id reuse)
        try:
            from .ops.modules import _SCRIPT_CLONE_KEEPALIVE
            _SCRIPT_CLONE_KEEPALIVE.clear()
        except Exception:
            pass
    logger.info("exported model to %s", path)
    return path


def release_port(ctx):
    """Release the port reserved for this node during bootstrap."""
    return ctx.release_port()


class DataFeed(object):
    """Consumer for RDD data fed by Spark tasks into this executor.

    Queue item protocol (the 'input' JoinableQueue carries *descriptors*; bulk
    rows live in the shared-memory ring):

    - ``None`` — end of feed
    - ``('end_partition',)`` — partition boundary marker
    - ``('shm', slot, nbytes, nrows)`` — pickled rows block in the ring
    - ``('rows', [row, ...])`` — inline block (no shared memory configured)
    """

    def __init__(self, mgr, train_mode=True, qname_in="input", qname_out="output",
                 input_mapping=None):
        self.mgr = mgr
        self.train_mode = train_mode
        self.qname_in = qname_in
        self.qname_out = qname_out
        self.done_feeding = False
        # Row values arrive in column-sorted order (the feeder selects columns
        # sorted by name, pipeline.py:147); bind tensor names in that same
        # order, not sorted by tensor name (reference TFNode.py:251).
        self.input_tensors = [t for _c, t in sorted(input_mapping.items())] \
            if input_mapping else None
        self._queue_in = mgr.get_queue(qname_in)
        self._queue_out = mgr.get_queue(qname_out)
        self._buffer = deque()
        self._ring = None

    def _attach_ring(self):
        if self._ring is None:
            from .utils import shmring
            name = self.mgr.get("ring_name")
            if name is None:
                raise RuntimeError("shm block received but no ring_name in manager kv")
            slots = self.mgr.get("ring_slots")
            slot_bytes = self.mgr.get("ring_slot_bytes")
            self._ring = shmring.BlockRing(
                name, slots, slot_bytes,
                data_queue=self._queue_in, free_queue=self.mgr.get_queue("free"),
                create=False)
        return self._ring

    def next_batch(self, batch_size):
        """Return up to ``batch_size`` rows (list of rows, or dict of columns
        when constructed with ``input_mapping``). May return fewer rows at
        end-of-feed or (inference mode) at a partition boundary."""
        batch = []
        while len(batch) < batch_size:
            if self._buffer:
                batch.append(self._buffer.popleft())
                continue
            item = self._queue_in.get(block=True)
            if item is None:
                self.done_feeding = True
                self._queue_in.task_done()
                break
            kind = item[0]
            if kind == "end_partition":
                self._queue_in.task_done()
                if not self.train_mode and len(batch) > 0:
                    break
                continue
            if kind == "shm":
                _, slot, nbytes, _nrows = item
                ring = self._attach_ring()
                import pickle
                rows = pickle.loads(ring.read(slot, nbytes))
                ring._free_q.put(slot)
                self._buffer.extend(rows)
                self._queue_in.task_done()
            elif kind == "rows":
                self._buffer.extend(item[1])
                self._queue_in.task_done()
            else:
                logger.warning("unknown queue item: %r", item)
                self._queue_in.task_done()
        if self.input_tensors is None:
            return batch
        # columnar dict keyed by sorted tensor name (reference TFNode.py:292-299)
        cols = {name: [] for name in self.input_tensors}
        for row in batch:
            for i, name in enumerate(self.input_tensors):
                cols[name].append(row[i])
        return cols

    def next_arrays(self):
        """Consume one array block fed via ``BlockRing.put_arrays``.

        Returns {name: np.ndarray} (views copied out of the ring), or None at
        end-of-feed. Partition markers are transparent here.
        """
        while True:
            item = self._queue_in.get(block=True)
            if item is None:
                self.done_feeding = True
                self._queue_in.task_done()
                return None
            kind = item[0]
            if kind == "end_partition":
                self._queue_in.task_done()
                continue
            if kind == "shm_arr":
                _, slot, _nbytes, meta = item
                ring = self._attach_ring()
                out = ring.read_arrays(slot, meta, copy=True)
                ring._free_q.put(slot)
                self._queue_in.task_done()
                return out
            raise RuntimeError(
                "next_arrays got a non-array item {!r}; mixed feeds should use "
                "next_batch".format(item[0]))

    def next_arrays_into(self, dest):
        """Like ``next_arrays`` but copies each named array straight into a
        caller-provided writable buffer (e.g. numpy views of *pinned* torch
        tensors) — one memcpy from shared memory to pinned staging, ready for
        hipMemcpyAsync. Returns True, or False at end-of-feed."""
        while True:
            item = self._queue_in.get(block=True)
            if item is None:
                self.done_feeding = True
                self._queue_in.task_done()
                return False
            kind = item[0]
            if kind == "end_partition":
                self._queue_in.task_done()
                continue
            if kind == "shm_arr":
                _, slot, _nbytes, meta = item
                ring = self._attach_ring()
                off = slot * ring.slot_bytes
                pos = 0
                for name, dtype, shape, n in meta:
                    buf = dest[name]
                    mv = memoryview(buf).cast("B")
                    mv[:n] = ring.shm.buf[off + pos:off + pos + n]
                    pos += n
                ring._free_q.put(slot)
                self._queue_in.task_done()
                return True
            raise RuntimeError(
                "next_arrays_into got non-array item {!r}".format(item[0]))

    def should_stop(self):
        return self.done_feeding

    def batch_results(self, results):
        """Push inference results 1:1 with consumed input rows."""
        for item in results:
            self._queue_out.put(item, block=True)

    def terminate(self):
        """Signal termination and drain any remaining feed items."""
        logger.info("DataFeed terminating: state -> terminating")
        self.mgr.set("state", "terminating")
        import queue as _q
        count = 0
        done = False
        while not done:
            try:
                item = self._queue_in.get(block=True, timeout=5)
                if item is not None and item[0] == "shm":
                    # return the slot so the producer never deadlocks on acquire
                    try:
                        self._attach_ring()._free_q.put(item[1])
                    except Exception:
                        pass
                self._queue_in.task_done()
                count += 1
            except _q.Empty:
                done = True
        logger.info("DataFeed drained %d items", count)


# Deprecated module-level functions from the TF1-era reference API surface
# (reference TFNode.py:157-159, 224-231) — kept as loud failures for parity.
def next_batch(mgr, batch_size, qname="input"):
    raise Exception("use TFNode.DataFeed instead")


def batch_results(mgr, results, qname="output"):
    raise Exception("use TFNode.DataFeed instead")


def terminate(mgr, qname="input"):
    raise Exception("use TFNode.DataFeed instead")
