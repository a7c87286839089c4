"""High-level cluster lifecycle API (parity: reference ``TFCluster.py``).

``run()`` turns N Spark executors into an N-node distributed PyTorch-ROCm
cluster: it assigns job roles (``ps``/``chief``/``evaluator``/``worker``), starts
the driver-side reservation server, and launches the per-executor bootstrap on a
background thread; ``train()``/``inference()`` feed RDDs through the
shared-memory ring; ``shutdown()`` tears everything down and surfaces trapped
worker errors.

Works against a real ``pyspark.SparkContext`` or any duck-typed context
providing ``parallelize``/``union``/``defaultParallelism`` (see
``tensorflowonspark_amd.local_context.LocalSparkContext`` for a Spark-free
multi-process stand-in used by tests and single-node deployments).
"""

import logging
import os
import random
import signal
import sys
import threading
import time

from . import TFSparkNode, reservation

logger = logging.getLogger(__name__)

# module-global status shared with the background start thread
# (reference TFCluster.py:40)
tf_status = {}


class InputMode(object):
    """TENSORFLOW = workers read data directly (TFRecord/files/synthetic);
    SPARK = RDD partitions are fed through the executors' shared-memory rings."""
    TENSORFLOW = 0
    SPARK = 1


class TFCluster(object):

    sc = None
    defaultFS = None
    working_dir = None
    num_executors = None
    nodeRDD = None
    cluster_id = None
    cluster_info = None
    cluster_meta = None
    input_mode = None
    queues = None
    server = None

    def train(self, dataRDD, num_epochs=0, feed_timeout=600, qname="input"):
        """Feed the data RDD to the cluster for training (InputMode.SPARK)."""
        logger.info("starting training")
        assert self.input_mode == InputMode.SPARK, "train() requires InputMode.SPARK"
        assert qname in self.queues, "unknown queue: {}".format(qname)

        if hasattr(dataRDD, "foreachRDD"):
            # Spark Streaming DStream
            dataRDD.foreachRDD(lambda rdd: rdd.foreachPartition(
                TFSparkNode.train(self.cluster_info, self.cluster_meta,
                                  feed_timeout=feed_timeout, qname=qname)))
            return

        if num_epochs == 0:
            num_epochs = 10  # reference default (TFCluster.py:88-93)
        rdds = [dataRDD] * num_epochs
        unionRDD = self.sc.union(rdds)
        unionRDD.foreachPartition(
            TFSparkNode.train(self.cluster_info, self.cluster_meta,
                              feed_timeout=feed_timeout, qname=qname))

    def inference(self, dataRDD, feed_timeout=600, qname="input",
                  qname_out="output"):
        """Feed an RDD for inference; returns an RDD of results (1:1 with rows)."""
        logger.info("starting inference")
        assert self.input_mode == InputMode.SPARK, "inference() requires InputMode.SPARK"
        assert qname in self.queues, "unknown queue: {}".format(qname)
        assert qname_out in self.queues, "unknown queue: {}".format(qname_out)
        return dataRDD.mapPartitions(
            TFSparkNode.inference(self.cluster_info, feed_timeout=feed_timeout,
                                  qname=qname, qname_out=qname_out))

    def shutdown(self, ssc=None, grace_secs=0, timeout=259200):
        """Stop the cluster; raise any error trapped in worker error queues.

        timeout: watchdog (SIGALRM) that cancels all jobs and exits if teardown
        wedges — default 3 days, parity with reference ``TFCluster.py:136-144``.
        """
        logger.info("waiting for cluster to shut down")
        workers = [n for n in self.cluster_info
                   if n["job_name"] in ("worker", "chief", "master")]
        ps_eval = [n for n in self.cluster_info
                   if n["job_name"] in ("ps", "evaluator")]

        timeout_handler_installed = False
        if threading.current_thread() is threading.main_thread() and hasattr(signal, "SIGALRM"):
            def _timeout_handler(signum, frame):
                logger.error("cluster shutdown timeout exceeded; exiting")
                try:
                    self.sc.cancelAllJobs()
                except Exception:
                    pass
                sys.exit(1)
            signal.signal(signal.SIGALRM, _timeout_handler)
            signal.alarm(timeout)
            timeout_handler_installed = True

        try:
            if ssc is not None:
                while not ssc.awaitTerminationOrTimeout(1):
                    if self.server.reservations.done():
                        logger.info("stop requested; stopping streaming context")
                        ssc.stop(stopSparkContext=False, stopGraceFully=True)

            if self.input_mode == InputMode.TENSORFLOW:
                # wait for all *worker*-role map_funs to finish (they read
                # data directly); ps/evaluator bootstrap tasks stay alive by
                # design until we stop them below — the reference polled
                # statusTracker for the same condition (TFCluster.py:154-169)
                target = len(workers)
                while (self.server.done_count() < target
                       and self._start_thread.is_alive()
                       and not tf_status.get("error")):
                    time.sleep(1)

            # push end-of-feed into worker queues
            if workers:
                workerRDD = self.sc.parallelize(
                    [n["executor_id"] for n in workers], len(workers))
                workerRDD.foreachPartition(
                    TFSparkNode.shutdown(self.cluster_info, self.queues,
                                         grace_secs=grace_secs))

            if tf_status.get("error"):
                logger.error("cluster error: %s", tf_status["error"])
                raise Exception("cluster startup/runtime error: {}".format(
                    tf_status["error"]))

            # driver-thread ps nodes stop in-process
            for srv in getattr(self, "_driver_ps", []):
                srv.stop()

            # stop executor ps/evaluator nodes: connect to their remote
            # managers from the driver and put None on 'control'
            # (reference TFCluster.py:186-194)
            from . import TFManager
            for node in ps_eval:
                if node.get("driver_ps"):
                    continue
                try:
                    m = TFManager.connect(tuple(node["addr"]),
                                          bytes.fromhex(node["authkey"]))
                    q = m.get_queue("control")
                    q.put(None, block=True)
                    q.join()
                except Exception as e:
                    logger.warning("couldn't stop %s node %d: %s",
                                   node["job_name"], node["executor_id"], e)

            # wait for the background start job to drain
            self._start_thread.join(timeout=60)
        finally:
            if timeout_handler_installed:
                signal.alarm(0)
            self.server.stop()
        logger.info("cluster shut down")

    def tensorboard_url(self):
        for node in self.cluster_info:
            if node.get("tb_port"):
                return "http://{}:{}".format(node["host"], node["tb_port"])
        return None


def run(sc, map_fun, tf_args, num_executors, num_ps=0, tensorboard=False,
        input_mode=InputMode.SPARK, log_dir=None, driver_ps_nodes=False,
        master_node="chief", reservation_timeout=600, queues=None,
        eval_node=False, release_port=True, num_gpus=1,
        ring_slots=8, ring_slot_bytes=8 << 20, block_rows=512):
    """Start a cluster across ``num_executors`` Spark executors.

    Role template (reference ``TFCluster.py:247-271``): executors [0, num_ps)
    are ``ps``; the next is ``master_node`` (chief) if set; the next is
    ``evaluator`` if ``eval_node``; the rest are ``worker``.
    """
    logger.info("Starting cluster: %d executors, %d ps, eval=%s",
                num_executors, num_ps, eval_node)
    queues = list(queues or ["input", "output", "error"])

    # -- role template --------------------------------------------------------
    cluster_template = {}
    executors = list(range(num_executors))
    if num_ps > 0:
        cluster_template["ps"] = executors[:num_ps]
        del executors[:num_ps]
    if master_node:
        cluster_template[master_node] = executors[:1]
        del executors[:1]
    if eval_node:
        cluster_template["evaluator"] = executors[:1]
        del executors[:1]
    if executors:
        cluster_template["worker"] = executors
    logger.info("cluster_template: %s", cluster_template)

    # -- defaultFS + working dir ---------------------------------------------
    defaultFS = "file://"
    try:
        hconf = sc._jsc.hadoopConfiguration()
        defaultFS = hconf.get("fs.defaultFS")
    except Exception:
        pass
    working_dir = os.getcwd()

    # -- reservation server ---------------------------------------------------
    server = reservation.Server(num_executors)
    server_addr = server.start()

    cluster_meta = {
        "id": random.getrandbits(64),
        "cluster_template": cluster_template,
        "num_executors": num_executors,
        "default_fs": defaultFS,
        "working_dir": working_dir,
        "server_addr": list(server_addr),
        "num_gpus": num_gpus,
        "release_port": release_port,
        "ring_slots": ring_slots,
        "ring_slot_bytes": ring_slot_bytes,
        "block_rows": block_rows,
    }

    tf_status.clear()

    # -- driver-side ps nodes (reference TFCluster.py:298-316) ----------------
    driver_ps = []
    executor_range = range(num_executors)
    if driver_ps_nodes and num_ps > 0:
        import threading as _threading

        from . import util as _util
        from .parallel import ps as _ps_mod
        host = _util.get_ip_address()
        if os.environ.get("TFOS_FORCE_LOOPBACK"):
            host = "127.0.0.1"
        for i, ps_id in enumerate(cluster_template.get("ps", [])):
            server_ps = _ps_mod.ParameterServer(port=0)
            t_ps = _threading.Thread(target=server_ps.serve_forever, daemon=True)
            t_ps.start()
            meta = {"executor_id": ps_id, "host": host, "job_name": "ps",
                    "task_index": i, "port": server_ps.port, "tb_pid": 0,
                    "tb_port": None, "addr": None, "authkey": "",
                    "driver_ps": True}
            server.reservations.add(meta)
            driver_ps.append(server_ps)
        executor_range = range(num_ps, num_executors)
        logger.info("running %d ps node(s) as driver threads", num_ps)

    # -- launch bootstrap job on a daemon thread ------------------------------
    nodeRDD = sc.parallelize(executor_range, len(executor_range))
    background = (input_mode == InputMode.SPARK)

    def _start(status):
        try:
            nodeRDD.foreachPartition(
                TFSparkNode.run(map_fun, tf_args, cluster_meta,
                                tensorboard=tensorboard, log_dir=log_dir,
                                queues=queues, background=background))
            status["done"] = True
        except Exception as e:
            logger.exception("cluster start job failed")
            status["error"] = str(e)

    t = threading.Thread(target=_start, args=(tf_status,), daemon=True)
    t.start()

    # -- wait for all reservations -------------------------------------------
    logger.info("waiting for %d reservations", num_executors)
    cluster_info = server.await_reservations(sc, tf_status, reservation_timeout)

    # duplicate (host, executor_id) detection from task retries
    # (reference TFCluster.py:357-372)
    seen = set()
    for node in cluster_info:
        key = (node["host"], node["executor_id"])
        if key in seen:
            raise Exception("duplicate node registration detected: {}".format(key))
        seen.add(key)
    logger.info("all nodes reserved: %s", cluster_info)

    cluster = TFCluster()
    cluster.sc = sc
    cluster.defaultFS = defaultFS
    cluster.working_dir = working_dir
    cluster.num_executors = num_executors
    cluster.nodeRDD = nodeRDD
    cluster.cluster_id = cluster_meta["id"]
    cluster.cluster_info = cluster_info
    cluster.cluster_meta = cluster_meta
    cluster.input_mode = input_mode
    cluster.queues = queues
    cluster.server = server
    cluster._start_thread = t
    cluster._driver_ps = driver_ps
    return cluster
