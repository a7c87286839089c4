"""Export->transform contract and the
DataFeed input_mapping ordering fix."""

import os

import pytest
import torch
import torch.nn as nn

from tensorflowonspark_amd import TFNode


class _Unscriptable(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc = nn.Linear(2, 1)

    def forward(self, x):
        # closures over self in a comprehension defeat torch.jit.script
        fns = [lambda t: self.fc(t)]
        return fns[0](x)


def test_export_falls_back_with_warning(tmp_path, caplog):
    import logging
    with caplog.at_level(logging.WARNING):
        path = TFNode.export_saved_model(_Unscriptable(), str(tmp_path))
    assert path.endswith("state_dict.pt")
    assert any("state_dict only" in r.message for r in caplog.records)


def test_export_require_script_raises(tmp_path):
    with pytest.raises(RuntimeError, match="TorchScript"):
        TFNode.export_saved_model(_Unscriptable(), str(tmp_path),
                                  require_script=True)


def test_export_scriptable_model(tmp_path):
    path = TFNode.export_saved_model(nn.Linear(3, 2), str(tmp_path))
    assert path.endswith("model.pt")
    m = torch.jit.load(path)
    assert m(torch.zeros(1, 3)).shape == (1, 2)


def test_fit_early_fails_on_state_dict_export(tmp_path, monkeypatch):
    """fit() must raise immediately when the chief exported only a state_dict
    (instead of letting transform() break later)."""
    from tensorflowonspark_amd import pipeline

    export_dir = tmp_path / "export"
    export_dir.mkdir()
    torch.save({}, str(export_dir / "state_dict.pt"))

    class _FakeCluster:
        def train(self, rdd, epochs):
            pass

        def shutdown(self, grace_secs=0):
            pass

    monkeypatch.setattr(pipeline.TFCluster, "run",
                        lambda *a, **k: _FakeCluster())

    class _FakeRDD:
        context = None

    class _FakeDF:
        sc = object()
        columns = ["x"]
        rdd = _FakeRDD()

        def select(self, cols):
            return self

    est = pipeline.TFEstimator(lambda args, ctx: None,
                               {"export_dir": str(export_dir)})
    with pytest.raises(RuntimeError, match="state_dict export"):
        est.fit(_FakeDF())


def test_datafeed_input_mapping_column_order():
    """Tensor binding must follow column-sorted order (reference
    TFNode.py:251), not tensor-name-sorted order: with mapping
    {'a': 'z_in', 'b': 'a_in'} column a's values bind to tensor z_in."""
    feed = TFNode.DataFeed.__new__(TFNode.DataFeed)
    feed.input_tensors = [t for _c, t in
                          sorted({"a": "z_in", "b": "a_in"}.items())]
    assert feed.input_tensors == ["z_in", "a_in"]
