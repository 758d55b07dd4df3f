"""Extended Isolation Forest (Hariri et al. 2018) estimator + model.

Reference: extended/ExtendedIsolationForest.scala:40-115 (estimator with
extensionLevel resolution) and extended/ExtendedIsolationForestModel.scala
(model). The resolved extensionLevel is set on the MODEL, never back on the
estimator (no leakage across fits — ExtendedIsolationForest.scala:102).
"""

from __future__ import annotations

import torch

from ..core import cpu_engine
from ..core.forest import ExtendedForest
from ..utils.params import ExtendedParams
from . import base
from .base import ModelBase
from .isolation_forest import IsolationForest


class ExtendedIsolationForest(IsolationForest):
    _params_cls = ExtendedParams
    _uid_prefix = "extended-isolation-forest"

    def _model_cls(self):
        return ExtendedIsolationForestModel

    def _build_local(self, X, rp, seed, t_lo, t_hi, n_local):
        T_local = t_hi - t_lo
        bag_idx = cpu_engine.sample_bags(
            n_local, T_local, rp.num_samples, seed,
            self.params.get("bootstrap"), tree_id_offset=t_lo,
        )
        feat_sub = cpu_engine.feature_subsets(
            rp.total_features, rp.num_features, T_local, seed, tree_id_offset=t_lo
        )
        if X.is_cuda and rp.num_samples <= 16384:
            from ..ops import gpu_engine

            return gpu_engine.build_extended_forest(
                X, bag_idx, feat_sub, seed, rp, tree_id_offset=t_lo
            )
        if X.is_cuda:
            import logging

            logging.getLogger(__name__).warning(
                "maxSamples %d exceeds the GPU build cap (16384); building "
                "trees on CPU (scoring stays on GPU)", rp.num_samples,
            )
            X = X.float().cpu()
        Xc = X.contiguous().float().numpy()
        return cpu_engine.build_extended_forest(
            Xc, bag_idx, feat_sub, seed, rp.num_samples, rp.num_features,
            rp.total_features, rp.extension_level, tree_id_offset=t_lo,
        )

    def _gather_forest(self, forest_local, comm, rp, total_features):
        if comm is None:
            return forest_local
        named = dict(
            feature=forest_local.feature,
            value=forest_local.value,
            right=forest_local.right,
            num_instances=forest_local.num_instances,
            node_count=forest_local.node_count,
            hyper_idx=forest_local.hyper_idx,
            hyper_w=forest_local.hyper_w,
            offset64=forest_local.offset64,
        )
        depth_np = getattr(forest_local, "depth_np", None)
        if depth_np is not None:
            named["depth_np"] = depth_np
        arrays = comm.all_gather_forest_arrays(**named)
        depth_np = arrays.pop("depth_np", None)
        forest = ExtendedForest(
            num_samples=forest_local.num_samples,
            num_features=forest_local.num_features,
            total_num_features=forest_local.total_num_features,
            extension_level=forest_local.extension_level,
            **arrays,
        )
        if depth_np is not None:
            forest.depth_np = depth_np
        return forest


class ExtendedIsolationForestModel(ModelBase):
    _uid_prefix = "extended-isolation-forest"

    def __init__(self, uid: str, forest: ExtendedForest, params: ExtendedParams):
        super().__init__(uid, params)
        self.forest = forest
        self._gpu_forest_cache = {}

    @property
    def num_samples(self) -> int:
        return self.forest.num_samples

    @property
    def num_features(self) -> int:
        return self.forest.num_features

    @property
    def total_num_features(self) -> int:
        return self.forest.total_num_features

    @property
    def extension_level(self) -> int:
        """The RESOLVED extension level this model was trained with
        (ExtendedIsolationForestModel requires it; :50-58)."""
        return self.forest.extension_level

    def _check_scorable(self):
        if self.forest.num_trees == 0:
            raise ValueError("this model has no trees; cannot transform")
        if self.forest.num_samples < 2:
            raise ValueError(
                f"numSamples is {self.forest.num_samples}; >= 2 required to transform"
            )

    def score(self, X: torch.Tensor) -> torch.Tensor:
        self._check_scorable()
        base.validate_feature_vector_size(self.forest.total_num_features, X.shape[1])
        if isinstance(X, torch.Tensor) and X.is_cuda:
            from ..ops import gpu_engine

            return gpu_engine.score_extended_forest(self, X)
        Xc = X.contiguous().float().numpy() if isinstance(X, torch.Tensor) else X
        return torch.from_numpy(cpu_engine.score_extended_forest(self.forest, Xc))

    def transform(self, data, comm=None):
        X, kind = base.extract_features(data, self.params.get("featuresCol"))
        base.check_output_columns(
            data, kind, self.params.get("scoreCol"), self.params.get("predictionCol")
        )
        scores = self.score(X)
        labels = self.labels_from_scores(scores)
        return base.attach_outputs(
            data, kind, self.params.get("scoreCol"),
            self.params.get("predictionCol"), scores, labels,
        )

    def __getattr__(self, attr):
        params = object.__getattribute__(self, "params")
        try:
            return getattr(params, attr)
        except AttributeError:
            raise AttributeError(attr)

    def save(self, path: str, overwrite: bool = False):
        from ..persist import model_io

        model_io.save_model(self, path, overwrite=overwrite)
        return self

    @property
    def write(self):
        from ..persist.model_io import WriteHandle

        return WriteHandle(self)

    @classmethod
    def load(cls, path: str) -> "ExtendedIsolationForestModel":
        from ..persist import model_io

        return model_io.load_model(path, expect_extended=True)
