#!/usr/bin/env python3
"""HTTP model serving over a cluster export.

The reference's serving story was JVM batch inference (Inference.scala) and
external TF Serving; this is the online-serving counterpart for this stack: a
FastAPI app over the chief's TorchScript export (``ctx.export_saved_model``),
running the model on the local MI355X when present.

  python examples/serving/serve.py --export_dir mnist_export --port 8000
  curl -X POST localhost:8000/predict -H 'content-type: application/json' \
       -d '{"inputs": [[0.1, 0.2, ...]], "shape": [1, 28, 28]}'
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def create_app(export_dir):
    import torch
    from fastapi import FastAPI
    from pydantic import BaseModel

    model = torch.jit.load(os.path.join(export_dir, "model.pt"),
                           map_location="cpu")
    device = "cuda" if torch.cuda.is_available() else "cpu"
    model = model.to(device)
    model.eval()

    class PredictRequest(BaseModel):
        inputs: list
        shape: list = []

    app = FastAPI(title="tfosr serving")

    @app.get("/health")
    def health():
        return {"status": "ok", "device": device}

    @app.post("/predict")
    def predict(req: PredictRequest):
        x = torch.as_tensor(req.inputs, dtype=torch.float32, device=device)
        if req.shape:
            x = x.reshape([x.shape[0]] + list(req.shape))
        with torch.no_grad():
            y = model(x)
        return {"outputs": y.cpu().tolist()}

    return app


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--export_dir", default="mnist_export")
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8000)
    args = p.parse_args()
    import uvicorn
    uvicorn.run(create_app(args.export_dir), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
