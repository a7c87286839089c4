"""Standalone C++ inference CLI (the reference's Inference.scala equivalent):
TorchScript export + TFRecords in, JSON predictions out, no Python involved."""

import os
import subprocess

import pytest
import torch

from tensorflowonspark_amd import tfrecord

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CLI = os.path.join(REPO, "tools", "bin", "tfosr_infer")


@pytest.fixture(scope="module")
def cli():
    if not os.path.exists(CLI):
        try:
            import sys
            sys.path.insert(0, REPO)
            from tools import build_ext
            build_ext.build()
            build_ext.build_infer_cli()
        except Exception as e:
            pytest.skip("could not build tfosr_infer: {}".format(e))
    return CLI


def test_cli_linear(tmp_path, cli):
    model = torch.nn.Linear(2, 1, bias=False)
    with torch.no_grad():
        model.weight.copy_(torch.tensor([[3.0, 2.0]]))
    export = tmp_path / "export"
    export.mkdir()
    torch.jit.script(model).save(str(export / "model.pt"))

    with tfrecord.TFRecordWriter(str(tmp_path / "data" / "part-r-00000")) as w:
        for row in ([1.0, 0.0], [0.0, 1.0], [2.0, 2.0]):
            w.write(tfrecord.encode_example({"x": row}))

    out = subprocess.run(
        [cli, "--export_dir", str(export), "--input", str(tmp_path / "data"),
         "--feature", "x"], capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    vals = [float(line.strip("[]")) for line in out.stdout.strip().splitlines()]
    assert vals == pytest.approx([3.0, 2.0, 10.0])


def test_cli_shaped_int_features(tmp_path, cli):
    """Int64 pixel features reshaped to an image tensor."""
    class Net(torch.nn.Module):
        def forward(self, x):
            return x.sum(dim=(1, 2, 3), keepdim=False).unsqueeze(1)

    export = tmp_path / "export"
    export.mkdir()
    torch.jit.script(Net()).save(str(export / "model.pt"))
    with tfrecord.TFRecordWriter(str(tmp_path / "d" / "part-r-00000")) as w:
        w.write(tfrecord.encode_example({"img": list(range(8))}))
    out = subprocess.run(
        [cli, "--export_dir", str(export), "--input", str(tmp_path / "d"),
         "--feature", "img", "--shape", "2,2,2"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    assert float(out.stdout.strip().strip("[]")) == pytest.approx(28.0)
