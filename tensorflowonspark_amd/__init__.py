"""tensorflowonspark_amd — MI355X-native Spark-orchestrated training/inference framework.

A from-scratch re-implementation of the capability set of yahoo/TensorFlowOnSpark
(reference v2.2.5) for AMD Instinct MI355X (gfx950) clusters:

* Each executor pins one MI355X and runs a PyTorch-ROCm worker process.
* Cluster bootstrap: TCP reservation/rendezvous server on the driver
  (capability parity with reference ``tensorflowonspark/reservation.py``), whose
  completed roster seeds ``torch.distributed`` process-group init over RCCL/xGMI.
* ``InputMode.SPARK``: RDD partitions flow through a shared-memory block ring into
  the GPU worker (replacing the reference's per-row pickled multiprocessing queues,
  reference ``TFSparkNode.py:500-502`` / ``TFNode.py:279``).
* Hot ops (fused BN+ReLU, softmax-cross-entropy, multi-tensor SGD/Adam, NHWC pack,
  MFMA GEMM) are hand-written HIP kernels for CDNA4 in ``csrc/``.

Public modules mirror the reference package layout so reference users can switch:
``TFCluster``, ``TFNode``, ``TFParallel``, ``TFManager``, ``reservation``,
``marker``, ``gpu_info``, ``util``, ``pipeline``, ``dfutil``, ``compat``.
"""

__version__ = "0.1.0"
