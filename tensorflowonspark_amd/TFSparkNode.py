"""Per-executor node runtime (parity: reference ``TFSparkNode.py``).

Bootstrap state machine (order is load-bearing, reference ``TFSparkNode.py:173-463``):
executor_id → GPU probe/pinning → role assignment → executor_id file → stale-manager
check → manager start (+ shared-memory ring) → port reservation → reservation
register/await → cluster_spec → context build → user fn launch.

MI355X-native differences from the reference:

* GPU probing uses ``rocm-smi``/``amd-smi`` (``gpu_info``) and exports
  ``HIP_VISIBLE_DEVICES`` + ``CUDA_VISIBLE_DEVICES`` (PyTorch-ROCm reads the
  latter).
* The reserved port on the chief seeds ``torch.distributed`` (RCCL over xGMI)
  instead of a TF gRPC server: ``ctx.init_process_group()`` computes
  rank/world_size from the completed reservation roster and rendezvouses on the
  chief's reserved port.
* Workers own a shared-memory block ring (``utils.shmring``) so feeder tasks move
  row *blocks*, not per-row pickles.
"""

import json
import logging
import multiprocessing
import os
import platform
import socket
import subprocess
import threading
import time
import traceback
import uuid

from . import TFManager, gpu_info, reservation, util

logger = logging.getLogger(__name__)


class TFNodeContext:
    """Encapsulates unique metadata for each node; passed to user ``map_fun``.

    Mirrors reference ``TFSparkNode.py:62-108`` and adds the torch-native
    surface: ``init_process_group``, ``device``, ``rank``/``world_size``.
    """

    def __init__(self, executor_id=0, job_name="", task_index=0, cluster_spec=None,
                 defaultFS="file://", working_dir=".", mgr=None, tmp_socket=None,
                 num_gpus=1):
        self.executor_id = executor_id
        self.job_name = job_name
        self.task_index = task_index
        self.cluster_spec = cluster_spec or {}
        self.defaultFS = defaultFS
        self.working_dir = working_dir
        self.mgr = mgr
        self.tmp_socket = tmp_socket
        self.num_gpus = num_gpus
        self._pg_initialized = False

    # -- parity helpers ------------------------------------------------------

    def absolute_path(self, path):
        from . import TFNode
        return TFNode.hdfs_path(self, path)

    def get_data_feed(self, train_mode=True, qname_in="input", qname_out="output",
                      input_mapping=None):
        from . import TFNode
        return TFNode.DataFeed(self.mgr, train_mode, qname_in, qname_out, input_mapping)

    def export_saved_model(self, model, export_dir):
        from . import TFNode
        return TFNode.export_saved_model(model, export_dir, is_chief=self.is_chief)

    def release_port(self):
        if self.tmp_socket is not None:
            try:
                self.tmp_socket.close()
            except OSError:
                pass
            self.tmp_socket = None

    # -- torch-native surface ------------------------------------------------

    @property
    def is_chief(self):
        return self.job_name in ("chief", "master") or (
            self.job_name == "worker" and "chief" not in self.cluster_spec
            and "master" not in self.cluster_spec and self.task_index == 0)

    def _world(self):
        """Ordered list of 'host:port' making up the torch.distributed world:
        chief/master first, then workers (sorted order is fixed by the roster)."""
        chief = self.cluster_spec.get("chief", self.cluster_spec.get("master", []))
        workers = self.cluster_spec.get("worker", [])
        return list(chief) + list(workers)

    @property
    def world_size(self):
        return len(self._world())

    @property
    def rank(self):
        me = "{}:{}".format(self._host, self._port) if hasattr(self, "_host") else None
        world = self._world()
        if me in world:
            return world.index(me)
        # fall back to role arithmetic
        if self.job_name in ("chief", "master"):
            return 0
        offset = 1 if ("chief" in self.cluster_spec or "master" in self.cluster_spec) else 0
        return self.task_index + offset

    @property
    def master_addr(self):
        world = self._world()
        host, port = world[0].rsplit(":", 1)
        return host, int(port)

    def init_process_group(self, backend=None, timeout_s=300):
        """Initialize torch.distributed across the worker+chief set.

        backend: 'nccl' (RCCL on ROCm) when a GPU is visible, else 'gloo'.
        The chief's reserved bootstrap port doubles as the rendezvous port (the
        placeholder socket is closed first).
        """
        import torch
        import torch.distributed as dist
        from datetime import timedelta
        if self._pg_initialized or dist.is_initialized():
            return dist
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        self.release_port()
        host, port = self.master_addr
        if os.environ.get("TFOS_FORCE_LOOPBACK_MASTER"):
            host = "127.0.0.1"
        dist.init_process_group(
            backend=backend,
            init_method="tcp://{}:{}".format(host, port),
            rank=self.rank, world_size=self.world_size,
            timeout=timedelta(seconds=timeout_s))
        self._pg_initialized = True
        if backend == "nccl":
            torch.cuda.set_device(0)  # each worker sees exactly one GPU
        logger.info("process group up: rank %d/%d backend=%s",
                    self.rank, self.world_size, backend)
        return dist

    @property
    def device(self):
        import torch
        return torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")

    # -- async (parameter-server) mode ---------------------------------------

    def run_parameter_server(self):
        """Body of a ``ps``-role ``map_fun``: serve parameter shards on this
        node's reserved port (the one advertised in cluster_spec['ps']) until
        the cluster shuts the node down."""
        assert self.job_name == "ps", "run_parameter_server on non-ps node"
        from .parallel import ps as ps_mod
        sock = self.tmp_socket
        self.tmp_socket = None
        server = ps_mod.ParameterServer(sock=sock)
        logger.info("parameter server %d serving on port %d",
                    self.task_index, server.port)
        server.serve_forever()

    def ps_client(self):
        """Worker-side client sharding buckets across the ps roster."""
        from .parallel import ps as ps_mod
        addrs = self.cluster_spec.get("ps", [])
        assert addrs, "cluster has no ps nodes (num_ps=0?)"
        return ps_mod.PSClient(addrs)

    # legacy parity
    def start_cluster_server(self, num_gpus=1, rdma=False):
        from . import TFNode
        return TFNode.start_cluster_server(self, num_gpus, rdma)


class TFSparkNode(object):
    """Per-python-worker singleton state (reference ``TFSparkNode.py:111-125``).

    ``owned_mgr`` pins the handle returned by ``TFManager.start`` for the life
    of the executor process: dropping the last reference to an *owning*
    BaseManager runs its finalizer, which terminates the manager server — a
    hazard when bootstrap and feeder tasks share one executor process."""
    mgr = None
    owned_mgr = None
    owned_ring = None
    cluster_id = None


def _get_manager(cluster_info, host, executor_id):
    """Reconnect to this executor's manager by (host, executor_id)."""
    for node in cluster_info:
        if node["host"] == host and node["executor_id"] == executor_id:
            addr = node["addr"]
            authkey = bytes.fromhex(node["authkey"])
            TFSparkNode.mgr = TFManager.connect(tuple(addr), authkey)
            break
    if TFSparkNode.mgr is None:
        raise Exception(
            "no TFManager found on {} for executor {}; cluster_info: {}".format(
                host, executor_id, cluster_info))
    state = str(TFSparkNode.mgr.get("state"))
    logger.info("connected to manager on %s executor=%d state=%s",
                host, executor_id, state)
    return TFSparkNode.mgr


def _has_spark_resource_api():
    try:
        from pyspark import TaskContext  # noqa: F401
        return hasattr(TaskContext, "resources")
    except ImportError:
        return False


def _get_gpus(num_gpus, worker_index=-1):
    """Resolve this node's GPU assignment, in priority order:
    (1) Spark 3 TaskContext resources, (2) K8s pod (granted by the scheduler),
    (3) rocm-smi/amd-smi probing (reference ``TFSparkNode.py:179-239``).
    Returns (gpu_string_or_None, from_probe): ``from_probe`` marks assignments
    that must be re-derived once the roster gives this node its deterministic
    index among same-host peers (second pass, reference ``:386-388``)."""
    if num_gpus == 0:
        return None, False
    if _has_spark_resource_api():
        try:
            from pyspark import TaskContext
            tc = TaskContext.get()
            if tc is not None:
                resources = tc.resources()
                if resources and "gpu" in resources:
                    return ",".join(resources["gpu"].addresses[:num_gpus]), False
        except Exception:
            pass
    if os.environ.get("SPARK_EXECUTOR_POD_IP"):
        # K8s: the pod was granted its GPUs by the scheduler; use them all
        ids = gpu_info._list_gpu_ids()
        if ids:
            return ",".join(str(i) for i in ids[:num_gpus]), False
        return None, False
    if gpu_info.is_gpu_available():
        return gpu_info.get_gpus(num_gpus, worker_index, format=str), True
    return None, False


def _host_peer_index(cluster_info, host, executor_id):
    """This node's rank among GPU-wanting nodes on the same host — the
    deterministic slice index for ``gpu_info.get_gpus`` (co-located executors
    must get disjoint GPUs; reference allocation-index math,
    ``TFSparkNode.py:213-228``)."""
    peers = sorted(n["executor_id"] for n in cluster_info
                   if n["host"] == host)
    return peers.index(executor_id)


def _get_cluster_spec(cluster_info):
    """Sorted-by-executor_id cluster spec {job_name: ['host:port', ...]}
    (reference ``TFSparkNode.py:46-59``)."""
    spec = {}
    for node in sorted(cluster_info, key=lambda n: n["executor_id"]):
        spec.setdefault(node["job_name"], []).append(
            "{}:{}".format(node["host"], node["port"]))
    return spec


def run(fn, tf_args, cluster_meta, tensorboard=False, log_dir=None, queues=None,
        background=False):
    """Factory: the once-per-executor bootstrap closure for foreachPartition."""

    def _mapfn(iterator):
        # one task per executor: consume the executor id
        for i in iterator:
            executor_id = i

        # -- role assignment --------------------------------------------------
        job_name = "worker"
        task_index = -1
        for jobtype, ids in cluster_meta["cluster_template"].items():
            if executor_id in ids:
                job_name = jobtype
                task_index = ids.index(executor_id)
                break
        assert task_index >= 0, "couldn't find executor_id in cluster_template"

        # -- GPU probe & pinning ---------------------------------------------
        num_gpus = int(cluster_meta.get("num_gpus", 1))
        wants_gpu = num_gpus > 0 and job_name in ("worker", "chief", "master", "evaluator")
        gpu_str, gpu_from_probe = _get_gpus(num_gpus if wants_gpu else 0)
        if gpu_str is not None:
            os.environ["HIP_VISIBLE_DEVICES"] = gpu_str
            os.environ["CUDA_VISIBLE_DEVICES"] = gpu_str
            logger.info("executor %d pinned to GPU(s) %s", executor_id, gpu_str)

        util.write_executor_id(executor_id)
        host = util.get_ip_address()
        if os.environ.get("TFOS_FORCE_LOOPBACK"):
            host = "127.0.0.1"
        cluster_id = cluster_meta["id"]

        # -- Spark-retry poisoning check (reference TFSparkNode.py:258-265) ---
        if TFSparkNode.mgr is not None and TFSparkNode.cluster_id == cluster_id:
            state = str(TFSparkNode.mgr.get("state"))
            if state != "stopped":
                raise Exception(
                    "TFManager for cluster {} already running on this executor "
                    "(state={}); Spark should retry elsewhere".format(cluster_id, state))

        # -- manager + ring ---------------------------------------------------
        authkey = uuid.uuid4().bytes
        if job_name in ("ps", "evaluator"):
            queue_names = ["control", "error"]
            mgr = TFManager.start(authkey, queue_names, "remote")
        else:
            queue_names = list(queues or ["input", "output", "error"])
            if "free" not in queue_names:
                queue_names.append("free")
            mgr = TFManager.start(authkey, queue_names, "local")
        mgr.set("state", "running")
        TFSparkNode.mgr = mgr
        TFSparkNode.owned_mgr = mgr
        TFSparkNode.cluster_id = cluster_id

        ring = None
        if job_name not in ("ps", "evaluator"):
            from .utils import shmring
            ring_name = "tfosr_{}_{}".format(cluster_id & 0xFFFFFFFF, executor_id)
            slots = int(cluster_meta.get("ring_slots", 8))
            slot_bytes = int(cluster_meta.get("ring_slot_bytes", 8 << 20))
            try:
                # clean any stale segment from a crashed prior run
                import multiprocessing.shared_memory as _shm
                try:
                    stale = _shm.SharedMemory(name=ring_name)
                    stale.close()
                    stale.unlink()
                except FileNotFoundError:
                    pass
                ring = shmring.BlockRing(ring_name, slots, slot_bytes,
                                         data_queue=mgr.get_queue("input"),
                                         free_queue=mgr.get_queue("free"),
                                         create=True)
                mgr.set("ring_name", ring_name)
                mgr.set("ring_slots", slots)
                mgr.set("ring_slot_bytes", slot_bytes)
                TFSparkNode.owned_ring = ring
            except Exception as e:
                logger.warning("shared-memory ring unavailable (%s); "
                               "falling back to inline row blocks", e)
                ring = None

        # -- optional TensorBoard-equivalent subprocess -----------------------
        tb_pid, tb_port = 0, None
        if tensorboard and job_name in ("worker", "chief", "master") and task_index == 0:
            tb_exec = util.find_in_path(os.environ.get("PATH", ""), "tensorboard")
            if tb_exec:
                tb_port = int(os.environ.get("TENSORBOARD_PORT", 0))
                if tb_port == 0:
                    s = socket.socket()
                    s.bind(("", 0))
                    tb_port = s.getsockname()[1]
                    s.close()
                logdir = log_dir or os.path.join(os.getcwd(), "tensorboard_logs")
                proc = subprocess.Popen(
                    [tb_exec, "--logdir", logdir, "--port", str(tb_port),
                     "--host", "0.0.0.0"])
                tb_pid = proc.pid
            else:
                logger.warning("tensorboard requested but not found in PATH")

        # -- port reservation + rendezvous ------------------------------------
        client = reservation.Client(cluster_meta["server_addr"])

        # idempotency for retried tasks (reference TFSparkNode.py:331-340)
        for node in client.get_reservations():
            if node["host"] == host and node["executor_id"] == executor_id:
                raise Exception(
                    "duplicate reservation for ({}, {}): Spark task retry "
                    "detected".format(host, executor_id))

        tmp_socket = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        tmp_socket.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        port_env = os.environ.get("TFOS_GRPC_PORT") or os.environ.get("TENSORFLOW_PORT")
        tmp_socket.bind(("", int(port_env) if port_env else 0))
        port = tmp_socket.getsockname()[1]

        node_meta = {
            "executor_id": executor_id,
            "host": host,
            "job_name": job_name,
            "task_index": task_index,
            "port": port,
            "tb_pid": tb_pid,
            "tb_port": tb_port,
            "addr": [host if job_name in ("ps", "evaluator") else "127.0.0.1",
                     mgr.address[1]],
            "authkey": authkey.hex(),
        }
        logger.info("registering node: %s", node_meta)
        client.register(node_meta)
        cluster_info = client.await_reservations()
        client.close()

        cluster_spec = _get_cluster_spec(cluster_info)

        # second GPU pass: with the roster known, probe-based assignments
        # become deterministic slices so co-located executors get disjoint
        # GPUs (reference TFSparkNode.py:386-388)
        if gpu_from_probe and wants_gpu:
            idx = _host_peer_index(cluster_info, host, executor_id)
            gpu_str = gpu_info.get_gpus(num_gpus, idx, format=str)
            os.environ["HIP_VISIBLE_DEVICES"] = gpu_str
            os.environ["CUDA_VISIBLE_DEVICES"] = gpu_str
            logger.info("executor %d re-pinned to GPU(s) %s (host index %d)",
                        executor_id, gpu_str, idx)
        # TF_CONFIG-equivalent for tooling that wants it
        os.environ["TFOS_CLUSTER_SPEC"] = json.dumps(
            {"cluster": cluster_spec,
             "task": {"type": job_name, "index": task_index}})

        ctx = TFNodeContext(
            executor_id=executor_id, job_name=job_name, task_index=task_index,
            cluster_spec=cluster_spec, defaultFS=cluster_meta["default_fs"],
            working_dir=cluster_meta["working_dir"], mgr=mgr,
            tmp_socket=tmp_socket, num_gpus=num_gpus)
        ctx._host = host
        ctx._port = port

        release_port = bool(cluster_meta.get("release_port", True))
        if release_port and not ctx.is_chief and job_name != "ps":
            # chief keeps its port (torch.distributed rendezvous, released
            # inside init_process_group); ps keeps its port (the parameter
            # service binds it in ctx.run_parameter_server)
            ctx.release_port()

        # -- launch user fn ----------------------------------------------------
        def wrapper_fn(args, context):
            if isinstance(args, list):
                import sys
                sys.argv = args
            fn(args, context)

        def wrapper_fn_background(args, context):
            try:
                wrapper_fn(args, context)
            except Exception:
                context.mgr.get_queue("error").put(traceback.format_exc())
                raise

        if job_name in ("ps", "evaluator") or background:
            p = multiprocessing.Process(target=wrapper_fn_background,
                                        args=(tf_args, ctx), daemon=True)
            if platform.system() == "Windows":
                raise Exception("background mode not supported on Windows")
            p.start()
            if job_name in ("ps", "evaluator"):
                # block until told to stop (None on 'control') or error
                queue = mgr.get_queue("control")
                equeue = mgr.get_queue("error")
                done = False
                while not done:
                    while queue.empty() and equeue.empty():
                        time.sleep(1)
                    if not equeue.empty():
                        e_str = equeue.get()
                        equeue.task_done()
                        raise Exception("exception in {}:\n{}".format(job_name, e_str))
                    msg = queue.get(block=True)
                    logger.info("%s got msg: %s", job_name, msg)
                    if msg is None:
                        done = True
                    queue.task_done()
                logger.info("%s node %d stopped", job_name, task_index)
        else:
            try:
                wrapper_fn(tf_args, ctx)
            except Exception:
                mgr.get_queue("error").put(traceback.format_exc())
                raise
            finally:
                # signal completion so shutdown() can distinguish "workers
                # finished" from "ps/evaluator still serving"
                try:
                    done_client = reservation.Client(cluster_meta["server_addr"])
                    done_client.notify_done(executor_id)
                    done_client.close()
                except Exception as e:
                    logger.debug("completion notify failed: %s", e)
            logger.info("%s node %d completed", job_name, task_index)

        return []

    return _mapfn


def _feed_partition(mgr, iterator, block_rows, qname="input"):
    """Pack an RDD partition into ring blocks (or inline blocks w/o shm)."""
    queue = mgr.get_queue(qname)
    ring = None
    ring_name = mgr.get("ring_name")
    if ring_name is not None:
        from .utils import shmring
        try:
            ring = shmring.BlockRing(
                str(ring_name), mgr.get("ring_slots"), mgr.get("ring_slot_bytes"),
                data_queue=queue, free_queue=mgr.get_queue("free"), create=False)
        except FileNotFoundError:
            ring = None
    import pickle
    count = 0
    block = []

    def flush():
        if not block:
            return
        if ring is not None:
            payload = pickle.dumps(block, protocol=pickle.HIGHEST_PROTOCOL)
            if len(payload) <= ring.slot_bytes:
                slot = ring.acquire()
                n = ring.write(slot, payload)
                queue.put(("shm", slot, n, len(block)))
                ring._free_q.task_done()
                del block[:]
                return
        queue.put(("rows", list(block)))
        del block[:]

    for row in iterator:
        block.append(row)
        count += 1
        if len(block) >= block_rows:
            flush()
    flush()
    if ring is not None:
        ring.close()
    return count


def train(cluster_info, cluster_meta, feed_timeout=600, qname="input"):
    """Factory: feeder closure for foreachPartition over the (unioned) data RDD."""

    def _train(iterator):
        host = util.get_ip_address()
        if os.environ.get("TFOS_FORCE_LOOPBACK"):
            host = "127.0.0.1"
        executor_id = util.read_executor_id()
        mgr = _get_manager(cluster_info, host, executor_id)
        queue = mgr.get_queue(qname)
        state = str(mgr.get("state"))
        terminating = state == "terminating"
        if terminating:
            # drain and count without feeding (reference TFSparkNode.py:492-496)
            count = sum(1 for _ in iterator)
            logger.info("terminating: skipped %d rows", count)
        else:
            block_rows = int(cluster_meta.get("block_rows", 512)) if cluster_meta else 512
            count = _feed_partition(mgr, iterator, block_rows, qname)
            logger.info("fed %d rows", count)
            # wait for the consumer to finish this partition, polling errors
            joined = [False]

            def _join():
                queue.join()
                joined[0] = True

            t = threading.Thread(target=_join, daemon=True)
            t.start()
            equeue = mgr.get_queue("error")
            timeout = feed_timeout
            while not joined[0]:
                time.sleep(1)
                if not equeue.empty():
                    e_str = equeue.get()
                    equeue.task_done()
                    raise Exception("exception in worker:\n" + e_str)
                timeout -= 1
                if timeout <= 0:
                    raise Exception("datafeed timed out after {}s awaiting "
                                    "consumption of partition".format(feed_timeout))
            state = str(mgr.get("state"))
            terminating = state == "terminating"
            if terminating:
                try:
                    client = reservation.Client(cluster_meta["server_addr"])
                    client.request_stop()
                    client.close()
                except Exception as e:
                    logger.debug("stop request failed (server may be gone): %s", e)
        return [terminating]

    return _train


def inference(cluster_info, feed_timeout=600, qname="input", qname_out="output"):
    """Factory: mapPartitions closure producing exactly one result per input row."""

    def _inference(iterator):
        host = util.get_ip_address()
        if os.environ.get("TFOS_FORCE_LOOPBACK"):
            host = "127.0.0.1"
        executor_id = util.read_executor_id()
        mgr = _get_manager(cluster_info, host, executor_id)
        queue_in = mgr.get_queue(qname)

        count = _feed_partition(mgr, iterator, 512, qname)
        queue_in.put(("end_partition",))
        if count == 0:
            # consume our own marker so join() can complete
            logger.info("empty partition")

        # wait for consumption w/ error polling
        joined = [False]

        def _join():
            queue_in.join()
            joined[0] = True

        t = threading.Thread(target=_join, daemon=True)
        t.start()
        equeue = mgr.get_queue("error")
        timeout = feed_timeout
        while not joined[0]:
            time.sleep(0.1)
            if not equeue.empty():
                e_str = equeue.get()
                equeue.task_done()
                raise Exception("exception in worker:\n" + e_str)
            timeout -= 0.1
            if timeout <= 0:
                raise Exception("datafeed timed out awaiting inference consumption")

        if count == 0:
            return []
        # pop exactly `count` results (reference invariant TFSparkNode.py:587-594)
        queue_out = mgr.get_queue(qname_out)
        results = []
        while len(results) < count:
            results.append(queue_out.get(block=True))
            queue_out.task_done()
        logger.info("collected %d inference results", len(results))
        return results

    return _inference


def shutdown(cluster_info, queues=("input",), grace_secs=0):
    """Factory: shutdown closure — push end-of-feed, surface trapped errors."""

    def _shutdown(iterator):
        host = util.get_ip_address()
        if os.environ.get("TFOS_FORCE_LOOPBACK"):
            host = "127.0.0.1"
        executor_id = util.read_executor_id()
        mgr = _get_manager(cluster_info, host, executor_id)

        # kill tensorboard if we spawned one
        for node in cluster_info:
            if node["host"] == host and node["executor_id"] == executor_id:
                if node.get("tb_pid"):
                    try:
                        os.kill(node["tb_pid"], 15)
                    except OSError:
                        pass

        logger.info("shutting down executor %d", executor_id)
        for q in queues:
            if q in ("error", "free"):
                continue  # end-of-feed goes to data queues only
            try:
                mgr.get_queue(q).put(None, block=True)
            except Exception as e:
                logger.warning("couldn't signal queue %s: %s", q, e)

        if grace_secs:
            time.sleep(grace_secs)

        # peek-and-requeue the error queue so Spark retries still observe it
        # (reference TFSparkNode.py:644-650)
        equeue = mgr.get_queue("error")
        if not equeue.empty():
            e_str = equeue.get()
            equeue.task_done()
            equeue.put(e_str)
            raise Exception("exception in worker:\n" + e_str)

        mgr.set("state", "stopped")
        return [True]

    return _shutdown
