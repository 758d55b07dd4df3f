"""Shared estimator/model plumbing (input handling, scoring, threshold).

The reference exposes spark.ml ``Estimator``/``Model`` pairs over Spark
DataFrames (IsolationForest.scala:46-105, IsolationForestModel.scala:116-151).
Here the data plane is tensors: ``fit``/``transform`` accept a 2-D
``numpy.ndarray`` or ``torch.Tensor`` [N, total_features] (float32/bf16;
a torch tensor on a ROCm device selects the HIP engine), or a pandas
DataFrame whose ``featuresCol`` column holds fixed-length vectors (the
spark.ml shape, kept for drop-in migrations). ``transform`` returns the
same kind of container with the score/prediction columns appended.
"""

from __future__ import annotations

import logging
import uuid
from typing import Optional, Tuple

import numpy as np
import torch

from ..core import threshold as threshold_mod
from ..utils.params import Params

logger = logging.getLogger(__name__)


def new_uid(prefix: str) -> str:
    return f"{prefix}_{uuid.uuid4().hex[:12]}"


def _is_pandas(data) -> bool:
    return type(data).__module__.startswith("pandas")


def extract_features(data, features_col: str) -> Tuple[torch.Tensor, object]:
    """Returns (float tensor [N, d] on the data's device, container kind).

    Mirrors Utils.validateAndTransformSchema's input checks
    (core/Utils.scala:35-65): the features column must exist and hold
    fixed-length numeric vectors.
    """
    if _is_pandas(data):
        if features_col not in data.columns:
            raise ValueError(f"features column {features_col!r} not found in DataFrame")
        mat = np.stack([np.asarray(v, dtype=np.float32) for v in data[features_col]])
        return torch.from_numpy(mat), "pandas"
    if isinstance(data, np.ndarray):
        if data.ndim != 2:
            raise ValueError(f"expected 2-D feature matrix, got shape {data.shape}")
        return torch.from_numpy(np.ascontiguousarray(data, dtype=np.float32)), "numpy"
    if isinstance(data, torch.Tensor):
        if data.dim() != 2:
            raise ValueError(f"expected 2-D feature matrix, got shape {tuple(data.shape)}")
        return data, "torch"
    raise TypeError(
        "fit/transform accept numpy.ndarray [N,d], torch.Tensor [N,d] or a "
        f"pandas DataFrame with a features column; got {type(data).__name__}"
    )


def check_output_columns(data, kind, score_col: str, prediction_col: str):
    """Output columns must not pre-exist (core/Utils.scala:46-58)."""
    if kind == "pandas":
        for col in (score_col, prediction_col):
            if col in data.columns:
                raise ValueError(f"output column {col!r} already exists")


def attach_outputs(data, kind, score_col, prediction_col, scores, labels):
    if kind == "pandas":
        out = data.copy()
        out[score_col] = np.asarray(scores)
        out[prediction_col] = np.asarray(labels)
        return out
    return {score_col: scores, prediction_col: labels}


def validate_feature_vector_size(total_num_features: int, d: int):
    """core/Utils.scala:67-72."""
    if total_num_features > 0 and d != total_num_features:
        raise ValueError(
            f"input feature vector length {d} does not match the "
            f"totalNumFeatures {total_num_features} this model was trained on"
        )


class ModelBase:
    """Shared model behavior: threshold handling and label emission."""

    def __init__(self, uid: str, params: Params):
        self.uid = uid
        self.params = params
        self._threshold: Optional[float] = None  # None == reference sentinel -1

    # -- threshold (IsolationForestModel.scala:85-96) -------------------
    @property
    def outlier_score_threshold(self) -> float:
        return -1.0 if self._threshold is None else self._threshold

    def set_outlier_score_threshold(self, value: float):
        if value == -1.0:
            self._threshold = None
            return self
        if not (0.0 <= value <= 1.0):
            raise ValueError(
                f"outlierScoreThreshold {value} must be in [0, 1] (or -1 for unset)"
            )
        self._threshold = float(value)
        return self

    # Spark-style aliases
    setOutlierScoreThreshold = set_outlier_score_threshold

    def getOutlierScoreThreshold(self) -> float:
        return self.outlier_score_threshold

    def labels_from_scores(self, scores):
        """predictedLabel = 1.0 iff score >= threshold; all 0.0 if unset
        (IsolationForestModel.scala:143-148)."""
        if isinstance(scores, torch.Tensor):
            if self._threshold is None:
                return torch.zeros_like(scores, dtype=torch.float64)
            return (scores.double() >= self._threshold).to(torch.float64)
        scores = np.asarray(scores)
        if self._threshold is None:
            return np.zeros(scores.shape, dtype=np.float64)
        return (scores.astype(np.float64) >= self._threshold).astype(np.float64)


def fit_threshold(model, scores: torch.Tensor, params: Params, comm=None):
    """computeAndSetModelThreshold (SharedTrainLogic.scala:175-242)."""
    contamination = params.get("contamination")
    if contamination == 0.0:
        logger.info("contamination is 0.0; outlier score threshold is not set")
        return
    err = params.get("contaminationError")
    th = threshold_mod.compute_threshold(scores, contamination, err, comm=comm)
    model.set_outlier_score_threshold(th)
    threshold_mod.verify_contamination(scores, th, contamination, err, comm=comm)
