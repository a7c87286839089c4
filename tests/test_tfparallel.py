"""TFParallel: independent per-executor instances
(shape parity: reference tests/test_TFParallel.py)."""

from tensorflowonspark_amd import TFParallel
from tensorflowonspark_amd.local_context import LocalSparkContext


def _fn(args, ctx):
    # each instance computes independently; no cluster, no collectives
    return {"executor": ctx.executor_id, "square": ctx.executor_id ** 2,
            "arg": args["k"]}


def test_parallel_run():
    sc = LocalSparkContext(num_executors=3)
    try:
        results = TFParallel.run(sc, _fn, {"k": 7}, 3, use_barrier=False)
        assert len(results) == 3
        assert sorted(r["executor"] for r in results) == [0, 1, 2]
        assert all(r["square"] == r["executor"] ** 2 for r in results)
        assert all(r["arg"] == 7 for r in results)
    finally:
        sc.stop()


def test_parallel_none_results_dropped():
    sc = LocalSparkContext(num_executors=2)
    try:
        results = TFParallel.run(sc, lambda a, c: None, None, 2,
                                 use_barrier=False)
        assert results == []
    finally:
        sc.stop()
