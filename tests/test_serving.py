"""HTTP serving example over a TorchScript export (FastAPI TestClient)."""

import importlib.util
import os

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_serving_predict(tmp_path):
    pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    model = torch.nn.Linear(4, 2, bias=False)
    with torch.no_grad():
        model.weight.copy_(torch.tensor([[1.0, 0, 0, 0], [0, 2.0, 0, 0]]))
    export = tmp_path / "export"
    export.mkdir()
    torch.jit.script(model).save(str(export / "model.pt"))

    spec = importlib.util.spec_from_file_location(
        "serve", os.path.join(REPO, "examples", "serving", "serve.py"))
    serve = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(serve)
    app = serve.create_app(str(export))
    client = TestClient(app)

    assert client.get("/health").json()["status"] == "ok"
    r = client.post("/predict", json={"inputs": [[1, 2, 3, 4], [5, 6, 7, 8]]})
    assert r.status_code == 200
    out = r.json()["outputs"]
    assert out == [[pytest.approx(1.0), pytest.approx(4.0)],
                   [pytest.approx(5.0), pytest.approx(12.0)]]
