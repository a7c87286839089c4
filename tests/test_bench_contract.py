"""bench.py driver contract: one JSON line from rank 0 with the required
fields, under both plain and torchrun invocation."""

import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _extract_json(stdout):
    lines = [ln for ln in stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, "exactly one JSON line expected: {!r}".format(lines)
    return json.loads(lines[0])


@pytest.mark.timeout(420)
def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "4", "--model", "resnet56_cifar"],
        capture_output=True, text=True, cwd=REPO, timeout=360)
    assert out.returncode == 0, out.stderr[-1500:]
    r = _extract_json(out.stdout)
    assert REQUIRED <= set(r.keys())
    assert r["n_gpus"] == 1 and r["steps"] == 2 and r["warmup"] == 1
    assert r["higher_is_better"] is True and r["scaling"] == "weak"
    assert r["unit"] == "images/sec" and r["value"] > 0
    assert r["dtype"] == "bf16" and "synthetic" in r["data"]
    assert r["config"]["global_batch"] == 4
    assert r["config"]["parallelism"] == "dp1"


@pytest.mark.timeout(500)
def test_bench_torchrun_world2_aggregates():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "4",
         "--model", "mnist_mlp"],
        capture_output=True, text=True, cwd=REPO, timeout=420)
    assert out.returncode == 0, out.stderr[-1500:]
    r = _extract_json(out.stdout)
    assert r["n_gpus"] == 2
    assert r["config"]["global_batch"] == 8       # whole-job aggregate
    assert r["config"]["parallelism"] == "dp2"


@pytest.mark.timeout(300)
def test_bench_no_launcher_does_not_hang():
    """`--gpus 8` without torchrun must run single-process (no peer wait)."""
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "8", "--steps", "1",
         "--warmup", "0", "--batch", "2", "--model", "mnist_mlp"],
        capture_output=True, text=True, cwd=REPO, timeout=240)
    assert out.returncode == 0, out.stderr[-1500:]
    assert _extract_json(out.stdout)["n_gpus"] == 1
