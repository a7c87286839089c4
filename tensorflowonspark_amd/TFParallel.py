"""Run a function as independent single-node instances on all executors
(parity: reference ``TFParallel.py:17-74`` — no cluster, no collectives; used
for embarrassingly-parallel batch inference, reference
``examples/mnist/keras/mnist_inference.py:79``).

Uses Spark barrier mode when available (gang-scheduling so co-located
instances can coordinate GPU placement); falls back to a plain job on
contexts without ``barrier()`` (e.g. the local executor pool).
"""

import logging

from . import gpu_info, util
from .TFSparkNode import TFNodeContext

logger = logging.getLogger(__name__)


def run(sc, map_fn, tf_args, num_executors, use_barrier=True):
    """Run ``map_fn(args, ctx)`` once per executor; returns collected results."""
    nodeRDD = sc.parallelize(range(num_executors), num_executors)

    def _run(it, peers=None):
        executor_id = None
        for i in it:
            executor_id = i
        # GPU placement: index among peers on this host
        num_gpus = getattr(tf_args, "num_gpus", 1) if tf_args is not None else 1
        worker_index = executor_id if peers else -1
        if num_gpus and gpu_info.is_gpu_available():
            util.single_node_env(num_gpus, worker_index)
        ctx = TFNodeContext(executor_id=executor_id, job_name="worker",
                            task_index=executor_id, cluster_spec={},
                            defaultFS="file://", working_dir=".", mgr=None,
                            num_gpus=num_gpus)
        result = map_fn(tf_args, ctx)
        return [result] if result is not None else []

    if use_barrier and hasattr(nodeRDD, "barrier"):
        def _barrier_run(it):
            from pyspark import BarrierTaskContext
            tc = BarrierTaskContext.get()
            peers = [info.address for info in tc.getTaskInfos()]
            tc.barrier()
            return _run(it, peers)
        return nodeRDD.barrier().mapPartitions(_barrier_run).collect()
    return nodeRDD.mapPartitions(lambda it: _run(it)).collect()
