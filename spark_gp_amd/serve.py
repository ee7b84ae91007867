"""Model serving: a FastAPI app over a saved model directory.

The reference has no serving story (models are not even persistable);
this serves the `save_model` format (`models/model_io.py`) over HTTP:

    python -m spark_gp_amd.serve /path/to/model --host 0.0.0.0 --port 8000

Endpoints:
* ``GET  /health``  -> {"status": "ok", "kind": ..., "m": ..., "d": ...}
* ``POST /predict`` body {"X": [[...], ...], "return_std": false}
  -> regression: {"mean": [...]} (+ "std" when requested)
  -> classification: {"proba": [...], "label": [...]}

Prediction cost is independent of the training-set size (PPA), so a
single GPU (or CPU) instance serves any model produced by any fit size;
batches stream through the HIP cross-kernel on MI355X.

NOTE: no `from __future__ import annotations` here — FastAPI must resolve
the function-local request model from live annotations.
"""

from typing import List, Optional

import numpy as np

from .models.classification import GaussianProcessClassificationModel
from .models.model_io import load_model
from .models.poisson import GaussianProcessPoissonModel


def create_app(model_path: str, device: str = "cpu"):
    """Build the FastAPI app (separate from __main__ so tests can use
    fastapi.testclient without spawning a server)."""
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    model = load_model(model_path, device=device)
    is_clf = isinstance(model, GaussianProcessClassificationModel)
    is_poisson = isinstance(model, GaussianProcessPoissonModel)
    active = model.raw.active_set
    m, d = active.shape

    class PredictRequest(BaseModel):
        X: List[List[float]]
        return_std: bool = False

    app = FastAPI(title="spark_gp_amd model server")

    @app.get("/health")
    def health():
        kind = ("classification" if is_clf
                else "poisson" if is_poisson else "regression")
        return {"status": "ok", "kind": kind,
                "m": int(m), "d": int(d), "device": device}

    @app.post("/predict")
    def predict(req: PredictRequest):
        X = np.asarray(req.X, dtype=np.float64)
        if X.ndim != 2 or X.shape[1] != d:
            raise HTTPException(
                status_code=422,
                detail=f"X must be [n, {d}] (model feature dim), "
                       f"got {list(X.shape)}")
        if is_clf:
            proba = model.predict_proba(X)[:, 1]
            return {"proba": proba.tolist(),
                    "label": (proba > 0.5).astype(int).tolist()}
        if is_poisson:
            mu, var = model.predict_latent(X)
            return {"rate": model.predict(X).tolist(),
                    "latent_mean": mu.tolist(),
                    "latent_var": var.tolist()}
        if req.return_std:
            mean, std = model.predict(X, return_std=True)
            return {"mean": mean.tolist(), "std": std.tolist()}
        return {"mean": model.predict(X).tolist()}

    return app


def main(argv: Optional[List[str]] = None):
    import argparse

    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("model_path")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--device", default=None,
                   help="'cpu' or 'cuda' (default: cuda when available)")
    args = p.parse_args(argv)

    import torch
    import uvicorn
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    uvicorn.run(create_app(args.model_path, device=device),
                host=args.host, port=args.port)


if __name__ == "__main__":
    main()
