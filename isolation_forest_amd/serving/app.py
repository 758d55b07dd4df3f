"""FastAPI scoring service over a persisted Isolation Forest model.

The model directory format is the Avro + metadata layout shared with the
reference (persist/model_io.py); the scorer is the engine's own
(CPU oracle or HIP kernels when the model is placed on a GPU device).
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch
from fastapi import FastAPI, HTTPException
from pydantic import BaseModel, Field

from ..persist import model_io


def load_any_model(path: str):
    """Load a saved model directory, standard or extended (auto-detect)."""
    meta = model_io._read_metadata(path)
    cls = meta.get("class", "")
    extended = "Extended" in cls.rsplit(".", 1)[-1]
    return model_io.load_model(path, expect_extended=extended)


class ScoreRequest(BaseModel):
    instances: List[List[float]] = Field(
        ..., description="batch of feature rows, shape [n, numFeatures]")


class ScoreResponse(BaseModel):
    scores: List[float]
    labels: Optional[List[int]] = None  # only when a threshold is set


def create_app(model_path: str, device: str = None) -> FastAPI:
    if device is None:
        device = "cuda:0" if torch.cuda.is_available() else "cpu"
    dev = torch.device(device)
    model = load_any_model(model_path)
    # dimension validation only when totalNumFeatures is known — legacy
    # models carry the -1 sentinel and accept any width, exactly like the
    # reference's transform (IsolationForestModel.scala:132-134)
    d = model.total_num_features if model.total_num_features > 0 else -1
    threshold = model.outlier_score_threshold

    app = FastAPI(title="isolation-forest-amd", version="1.0")
    app.state.model = model
    app.state.device = str(dev)

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    @app.get("/v1/model")
    def model_info():
        return {
            "class": type(model).__name__,
            "numFeatures": model.num_features,
            "totalNumFeatures": model.total_num_features,
            "numSamples": model.num_samples,
            "numTrees": model.forest.num_trees,
            "outlierScoreThreshold": threshold,
            "device": str(dev),
        }

    @app.post("/v1/score", response_model=ScoreResponse)
    def score(req: ScoreRequest):
        if not req.instances:
            return ScoreResponse(scores=[])
        try:
            X = np.asarray(req.instances, dtype=np.float32)
        except ValueError:
            raise HTTPException(
                status_code=400, detail="instances must be a rectangular "
                "batch of numeric rows")
        if X.ndim != 2 or (d > 0 and X.shape[1] != d):
            raise HTTPException(
                status_code=400,
                detail=f"each instance must have {d} features, "
                       f"got shape {list(X.shape)}")
        try:
            with torch.no_grad():
                scores = model.score(torch.from_numpy(X).to(dev))
        except (IndexError, RuntimeError, ValueError) as e:
            # legacy models (no totalNumFeatures) skip up-front width
            # validation, matching the reference; a width the trees cannot
            # index still fails cleanly here
            raise HTTPException(status_code=400,
                                detail=f"scoring failed: {e}")
        scores_np = scores.float().cpu().numpy()
        labels = None
        if threshold >= 0:
            labels = (scores_np >= threshold).astype(int).tolist()
        return ScoreResponse(scores=scores_np.tolist(), labels=labels)

    return app
