"""End-to-end example smoke runs (subprocess, tiny configs) — these are the
user-facing entry points; keep them green."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, *args, timeout=420):
    return subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", script)] + list(args),
        capture_output=True, text=True, timeout=timeout, cwd=REPO)


@pytest.mark.timeout(500)
def test_mnist_spark_example(tmp_path):
    r = _run("mnist/mnist_spark.py", "--cluster_size", "2", "--epochs", "1",
             "--num_gpus", "0", "--batch_size", "128",
             "--data", str(tmp_path / "d" / "mnist.csv"),
             "--model_dir", str(tmp_path / "m"),
             "--export_dir", str(tmp_path / "e"))
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(str(tmp_path / "e" / "model.pt"))


@pytest.mark.timeout(500)
def test_resnet_async_example():
    r = _run("resnet/resnet_async.py", "--cluster_size", "3", "--num_ps", "1",
             "--records", "48", "--batch_size", "8", "--num_gpus", "0")
    assert r.returncode == 0, r.stderr[-2000:]
    assert "async run complete" in r.stdout


@pytest.mark.timeout(500)
def test_mnist_eval_example(tmp_path):
    """Evaluator-role example (reference estimator/mnist_tf.py eval_node
    flow): a dedicated evaluator polls checkpoints and writes eval events."""
    import glob
    r = _run("mnist/mnist_eval.py", "--cluster_size", "3", "--epochs", "1",
             "--batch_size", "64", "--ckpt_every", "3", "--eval_timeout", "15",
             "--data", str(tmp_path / "d" / "mnist.csv"),
             "--model_dir", str(tmp_path / "m"))
    assert r.returncode == 0, r.stderr[-2000:]
    # evaluator output lives in the executor process; judge by its artifacts
    evs = glob.glob(str(tmp_path / "m" / "eval" / "events.out.tfevents.*"))
    assert evs, "no eval event files written"
    from tensorflowonspark_amd import tfrecord
    from tensorflowonspark_amd.utils import events
    recs = list(tfrecord.tfrecord_iterator(evs[0], verify=True))
    assert len(recs) >= 2
    _, step, sc = events.decode_scalar_event(recs[-1])
    assert "eval_acc" in sc
