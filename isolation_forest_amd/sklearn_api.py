"""scikit-learn-style adapter.

The reference plugs into spark.ml Pipelines (Estimator/Model); the
pythonic analog is the sklearn contract. This thin wrapper maps our
engine onto sklearn's ``IsolationForest`` surface so it drops into
sklearn pipelines/metrics without adapters:

* ``fit(X, y=None)`` / ``fit_predict(X)``
* ``score_samples(X)``   = NEGATIVE outlier score (sklearn convention:
  higher = more normal; ours is in (0,1), higher = more anomalous)
* ``decision_function``  = score_samples + threshold (negative = outlier)
* ``predict(X)``         = -1 outlier / +1 inlier

Contamination follows sklearn semantics ("auto" => the reference default
of no threshold; then predict uses the conventional 0.5 score cut).
"""

from __future__ import annotations

from typing import Optional, Union

import numpy as np
import torch

from .models.isolation_forest import IsolationForest as _Engine


class IsolationForestSKL:
    """sklearn-compatible front end over the MI355X engine."""

    def __init__(self, n_estimators: int = 100,
                 max_samples: Union[int, float, str] = 256,
                 contamination: Union[float, str] = "auto",
                 max_features: Union[int, float] = 1.0,
                 bootstrap: bool = False,
                 random_state: Optional[int] = None,
                 device: Optional[str] = None):
        self.n_estimators = n_estimators
        self.max_samples = max_samples
        self.contamination = contamination
        self.max_features = max_features
        self.bootstrap = bootstrap
        self.random_state = random_state
        self.device = device

    # -- sklearn plumbing ------------------------------------------------
    def get_params(self, deep: bool = True):
        return {
            "n_estimators": self.n_estimators,
            "max_samples": self.max_samples,
            "contamination": self.contamination,
            "max_features": self.max_features,
            "bootstrap": self.bootstrap,
            "random_state": self.random_state,
            "device": self.device,
        }

    def set_params(self, **params):
        for k, v in params.items():
            if not hasattr(self, k):
                raise ValueError(f"invalid parameter {k!r}")
            setattr(self, k, v)
        return self

    # -- core ------------------------------------------------------------
    def _to_tensor(self, X):
        if isinstance(X, torch.Tensor):
            t = X
        else:
            t = torch.from_numpy(np.ascontiguousarray(X, dtype=np.float32))
        if self.device is not None:
            t = t.to(self.device)
        return t

    def fit(self, X, y=None):
        contamination = self.contamination
        auto = contamination == "auto"
        X_t = self._to_tensor(X)
        n_rows = int(X_t.shape[0])
        ms = self.max_samples
        if ms == "auto":
            ms = 256
        if not isinstance(ms, float) or ms > 1.0:
            # sklearn contract: integer/"auto" max_samples is clamped to
            # n_samples (sklearn's IsolationForest uses min(256, n)); the
            # engine would otherwise reject maxSamples > row count.
            ms = min(int(ms), n_rows)
        est = _Engine(
            numEstimators=int(self.n_estimators),
            maxSamples=float(ms),
            contamination=0.0 if auto else float(contamination),
            contaminationError=0.0 if auto else float(contamination) * 0.1,
            maxFeatures=float(self.max_features),
            bootstrap=bool(self.bootstrap),
            randomSeed=int(self.random_state)
            if self.random_state is not None else 1,
        )
        self.model_ = est.fit(X_t)
        if auto:
            # conventional score cut at 0.5 (Liu et al.: scores above 0.5
            # indicate anomalies)
            self.offset_ = -0.5
        else:
            self.offset_ = -float(self.model_.outlier_score_threshold)
        return self

    def score_samples(self, X) -> np.ndarray:
        scores = self.model_.score(self._to_tensor(X))
        return -scores.float().cpu().numpy()

    def decision_function(self, X) -> np.ndarray:
        return self.score_samples(X) - self.offset_

    def predict(self, X) -> np.ndarray:
        return np.where(self.decision_function(X) < 0, -1, 1)

    def fit_predict(self, X, y=None) -> np.ndarray:
        return self.fit(X).predict(X)
