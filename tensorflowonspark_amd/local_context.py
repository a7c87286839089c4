"""A Spark-free stand-in for SparkContext: N executor *processes*, 1 core each.

The reference framework requires executors to be separate OS processes ("TFoS
assumes that the executors run in separate processes", reference
``tests/README.md:10``) and simulated multi-node with a 2-worker Spark
Standalone cluster. This module provides the same shape without a JVM: a
:class:`LocalSparkContext` duck-types the subset of the SparkContext/RDD API the
framework uses (``parallelize``, ``union``, ``foreachPartition``,
``mapPartitions``/``collect``), backed by persistent executor processes, each
with its own working directory (so the ``executor_id`` file protocol works
exactly as on a real cluster).

Task routing mirrors Spark scheduling where it matters:
* the *bootstrap* job pins partition i to executor i (``pin_to_executor``);
* all other jobs dispatch partitions to whichever executor is idle — ps and
  evaluator executors stay busy inside their bootstrap task, so feeder and
  shutdown tasks naturally land on worker executors, as on real Spark.

Jobs may run concurrently from multiple driver threads (the cluster-start job
runs on a daemon thread while feed jobs run on the main thread, mirroring
reference ``TFCluster.py:318-336``); a single dispatcher thread owns executor
assignment and completion routing.

This also serves production single-node use: 8 executors = 8 MI355X on one box
without a Spark install.
"""

import logging
import multiprocessing

import cloudpickle
import os
import queue as pyqueue
import shutil
import tempfile
import threading
import traceback
import uuid

logger = logging.getLogger(__name__)

_STOP = "__stop__"


def _index_wrap(fn, idx, it):
    return fn(idx, it)


def _executor_main(exec_id, workdir, task_q, result_q, env):
    import cloudpickle
    os.makedirs(workdir, exist_ok=True)
    os.chdir(workdir)
    os.environ.update(env)
    while True:
        task = task_q.get()
        if task == _STOP:
            # shut down any manager server this executor owns so no orphan
            # process outlives the context (it would hold stdio pipes open)
            try:
                from .TFSparkNode import TFSparkNode
                if TFSparkNode.owned_ring is not None:
                    TFSparkNode.owned_ring.close()
                    TFSparkNode.owned_ring.unlink()
                if TFSparkNode.owned_mgr is not None:
                    TFSparkNode.owned_mgr.shutdown()
            except Exception:
                pass
            break
        job_id, pid, fn, data, collect = cloudpickle.loads(task)
        try:
            out = fn(iter(data))
            payload = list(out) if collect and out is not None else None
            result_q.put((job_id, pid, exec_id, "ok", payload))
        except BaseException:
            result_q.put((job_id, pid, exec_id, "err", traceback.format_exc()))


class LocalRDD:
    def __init__(self, sc, partitions):
        self.sc = sc
        self.partitions = [list(p) for p in partitions]

    def getNumPartitions(self):
        return len(self.partitions)

    def foreachPartition(self, fn, pin_to_executor=False):
        self.sc._run_job(self.partitions, fn, collect=False, pin=pin_to_executor)

    def mapPartitions(self, fn):
        return _LazyRDD(self.sc, self.partitions, fn)

    def mapPartitionsWithIndex(self, fn):
        return _LazyRDD(self.sc, self.partitions, fn, with_index=True)

    def map(self, fn):
        return _LazyRDD(self.sc, self.partitions,
                        lambda it: (fn(x) for x in it))

    def collect(self):
        return [x for p in self.partitions for x in p]

    def count(self):
        return sum(len(p) for p in self.partitions)


class _LazyRDD:
    """mapPartitions result; executes on .collect()/.foreachPartition()."""

    def __init__(self, sc, partitions, fn, with_index=False):
        self.sc = sc
        self.partitions = partitions
        self.fn = fn
        self.with_index = with_index

    def getNumPartitions(self):
        return len(self.partitions)

    def mapPartitions(self, fn2):
        prev = self.fn

        def chained(it):
            return fn2(iter(list(prev(it))))
        return _LazyRDD(self.sc, self.partitions, chained)

    def collect(self):
        results = self.sc._run_job(self.partitions, self.fn, collect=True,
                                   with_index=self.with_index)
        return [x for p in results for x in (p or [])]

    def foreachPartition(self, fn2):
        prev = self.fn

        def chained(it):
            fn2(iter(list(prev(it))))
            return []
        self.sc._run_job(self.partitions, chained, collect=False)


def _infer_dtype(v):
    if isinstance(v, bool):
        return "boolean"
    if isinstance(v, int):
        return "bigint"
    if isinstance(v, float):
        return "double"
    if isinstance(v, str):
        return "string"
    if isinstance(v, (bytes, bytearray)):
        return "binary"
    if isinstance(v, (list, tuple)):
        inner = _infer_dtype(v[0]) if v else "double"
        return "array<{}>".format(inner)
    return "string"


class LocalDataFrame:
    """Minimal Spark-SQL-DataFrame stand-in: named, typed columns over rows.

    Supports the subset the pipeline/dfutil layers use: ``columns``,
    ``dtypes``, ``select``, ``.rdd``, ``collect``, ``count``.
    """

    def __init__(self, sc, rows, columns, dtypes=None):
        self.sc = sc
        self._rows = [tuple(r) for r in rows]
        self.columns = list(columns)
        if dtypes is None:
            first = self._rows[0] if self._rows else tuple("" for _ in columns)
            dtypes = [_infer_dtype(v) for v in first]
        self.dtypes = list(zip(self.columns, dtypes))

    def select(self, *cols):
        if len(cols) == 1 and isinstance(cols[0], (list, tuple)):
            cols = list(cols[0])
        idx = [self.columns.index(c) for c in cols]
        rows = [tuple(r[i] for i in idx) for r in self._rows]
        dt = [self.dtypes[i][1] for i in idx]
        return LocalDataFrame(self.sc, rows, list(cols), dt)

    @property
    def rdd(self):
        return self.sc.parallelize(self._rows, self.sc.defaultParallelism)

    def collect(self):
        return list(self._rows)

    def count(self):
        return len(self._rows)

    def show(self, n=20):
        print(self.columns)
        for r in self._rows[:n]:
            print(r)


class _Job:
    def __init__(self, job_id, partitions, fn, collect, with_index=False):
        self.job_id = job_id
        self.partitions = partitions
        self.fn = fn
        self.collect = collect
        self.with_index = with_index
        self.pending = list(range(len(partitions)))
        self.results = [None] * len(partitions)
        self.done = 0
        self.error = None
        self.event = threading.Event()

    def task_fn(self, pid):
        import functools
        if self.with_index:
            return functools.partial(_index_wrap, self.fn, pid)
        return self.fn


class LocalDStream:
    """Minimal DStream stand-in: RDDs pushed via the owning streaming context
    flow to every registered foreachRDD callback."""

    def __init__(self, ssc):
        self.ssc = ssc
        self._callbacks = []

    def foreachRDD(self, fn):
        self._callbacks.append(fn)


class LocalStreamingContext:
    """Spark-Streaming stand-in for the DStream feed path
    (reference streaming example: ``mnist_spark_streaming.py``).

    ``queueStream()`` returns a stream; ``push(rdd)`` delivers a micro-batch
    to the registered callbacks on a worker thread; ``stop()`` /
    ``awaitTerminationOrTimeout`` mirror the pyspark surface TFCluster uses.
    """

    def __init__(self, sc):
        self.sc = sc
        self._queue = pyqueue.Queue()
        self._streams = []
        self._stopped = threading.Event()
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def queueStream(self):
        s = LocalDStream(self)
        self._streams.append(s)
        return s

    def push(self, rdd):
        self._queue.put(rdd)

    def _loop(self):
        while not self._stopped.is_set():
            try:
                rdd = self._queue.get(timeout=0.2)
            except pyqueue.Empty:
                continue
            for stream in self._streams:
                for cb in stream._callbacks:
                    try:
                        cb(rdd)
                    except Exception:
                        logger.exception("streaming callback failed")

    def awaitTerminationOrTimeout(self, timeout):
        return self._stopped.wait(timeout)

    def stop(self, stopSparkContext=False, stopGraceFully=True):
        if stopGraceFully:
            # drain pending micro-batches first
            while not self._queue.empty() and not self._stopped.is_set():
                import time as _t
                _t.sleep(0.1)
        self._stopped.set()
        if stopSparkContext:
            self.sc.stop()


class LocalSparkContext:
    """N persistent executor processes with Spark-like task dispatch."""

    def __init__(self, num_executors=2, workdir_root=None, env=None):
        self.num_executors = num_executors
        self.defaultParallelism = num_executors
        self._root = workdir_root or tempfile.mkdtemp(prefix="tfos_local_")
        self._own_root = workdir_root is None
        # fork context: executors inherit imports cheaply and closures don't
        # need spawn-compatible __main__. Create the context BEFORE running
        # torch compute in the driver — forking after OpenMP/ATen thread
        # pools exist is the classic libgomp post-fork deadlock. (Real Spark
        # executors are separate JVM-launched processes and are unaffected.)
        ctx = multiprocessing.get_context("fork")
        self._result_q = ctx.Queue()
        self._task_qs = []
        self._procs = []
        # cap math-library threads: N executor processes each defaulting to
        # all cores oversubscribes the host badly during CPU runs
        base_env = {"TFOS_FORCE_LOOPBACK": "1", "TFOS_FORCE_LOOPBACK_MASTER": "1",
                    "OMP_NUM_THREADS": os.environ.get("OMP_NUM_THREADS", "4"),
                    "MKL_NUM_THREADS": os.environ.get("MKL_NUM_THREADS", "4")}
        base_env.update(env or {})
        for i in range(num_executors):
            tq = ctx.Queue()
            # NOT daemonic: executor tasks spawn children (the TFManager server
            # process and background worker processes), which daemonic
            # processes are forbidden to do.
            p = ctx.Process(
                target=_executor_main,
                args=(i, os.path.join(self._root, "executor_{}".format(i)),
                      tq, self._result_q, base_env),
                daemon=False)
            p.start()
            self._task_qs.append(tq)
            self._procs.append(p)
        self._busy = [0] * num_executors
        self._lock = threading.Lock()
        self._jobs = {}        # job_id -> _Job (with unpinned pending work)
        self._job_order = []   # FIFO of job_ids with pending partitions
        self._stopped = False
        self._dispatcher = threading.Thread(target=self._dispatch_loop, daemon=True)
        self._dispatcher.start()
        import atexit
        atexit.register(self.stop)

    # -- SparkContext surface -------------------------------------------------

    def parallelize(self, seq, numSlices=None):
        seq = list(seq)
        n = numSlices or self.defaultParallelism
        n = max(1, min(n, max(1, len(seq))))
        size = len(seq) // n
        extra = len(seq) % n
        parts, start = [], 0
        for i in range(n):
            end = start + size + (1 if i < extra else 0)
            parts.append(seq[start:end])
            start = end
        return LocalRDD(self, parts)

    def union(self, rdds):
        parts = []
        for r in rdds:
            parts.extend(r.partitions)
        return LocalRDD(self, parts)

    def createDataFrame(self, data, columns, dtypes=None):
        return LocalDataFrame(self, data, columns, dtypes)

    def cancelAllJobs(self):
        pass

    def stop(self):
        if self._stopped:
            return
        self._stopped = True
        for tq in self._task_qs:
            try:
                tq.put(_STOP)
            except Exception:
                pass
        for p in self._procs:
            p.join(timeout=5)
            if p.is_alive():
                p.terminate()
        if self._own_root:
            shutil.rmtree(self._root, ignore_errors=True)

    # -- scheduler ------------------------------------------------------------

    def _run_job(self, partitions, fn, collect, pin=False, timeout=None,
                 with_index=False):
        """Submit one task per partition; block until all complete or one fails."""
        if self._stopped:
            raise RuntimeError("context is stopped")
        job = _Job(uuid.uuid4().hex, partitions, fn, collect, with_index)
        with self._lock:
            self._jobs[job.job_id] = job
            if pin:
                assert len(partitions) <= self.num_executors, \
                    "pinned job needs {} executors, have {}".format(
                        len(partitions), self.num_executors)
                for pid in list(job.pending):
                    self._task_qs[pid].put(cloudpickle.dumps(
                        (job.job_id, pid, job.task_fn(pid), partitions[pid],
                         collect)))
                    self._busy[pid] += 1
                job.pending = []
            else:
                self._job_order.append(job.job_id)
        if not job.event.wait(timeout=timeout):
            raise TimeoutError("job timed out")
        with self._lock:
            self._jobs.pop(job.job_id, None)
        if job.error is not None:
            raise RuntimeError(job.error)
        return job.results

    def _dispatch_loop(self):
        while not self._stopped:
            with self._lock:
                self._assign_pending()
            try:
                msg = self._result_q.get(timeout=0.2)
            except pyqueue.Empty:
                for i, p in enumerate(self._procs):
                    if not p.is_alive() and not self._stopped:
                        self._fail_all("executor {} died (exitcode {})".format(
                            i, p.exitcode))
                        return
                continue
            job_id, pid, eid, status, payload = msg
            with self._lock:
                self._busy[eid] = max(0, self._busy[eid] - 1)
                job = self._jobs.get(job_id)
                if job is None:
                    continue
                if status == "err":
                    job.error = "task {} failed on executor {}:\n{}".format(
                        pid, eid, payload)
                    job.event.set()
                    continue
                job.results[pid] = payload
                job.done += 1
                if job.done == len(job.partitions):
                    job.event.set()

    def _assign_pending(self):
        """Assign queued unpinned partitions to idle executors (lock held)."""
        idle = [i for i, b in enumerate(self._busy) if b == 0]
        while idle and self._job_order:
            jid = self._job_order[0]
            job = self._jobs.get(jid)
            if job is None or not job.pending:
                self._job_order.pop(0)
                continue
            eid = idle.pop()
            pid = job.pending.pop(0)
            self._task_qs[eid].put(cloudpickle.dumps(
                (job.job_id, pid, job.task_fn(pid), job.partitions[pid],
                 job.collect)))
            self._busy[eid] += 1
            if not job.pending:
                self._job_order.pop(0)

    def _fail_all(self, msg):
        with self._lock:
            for job in self._jobs.values():
                job.error = msg
                job.event.set()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.stop()
