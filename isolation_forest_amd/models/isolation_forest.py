"""Standard Isolation Forest estimator + model.

Reference: IsolationForest.scala (estimator, :46-105) and
IsolationForestModel.scala (model, :116-151). The fit pipeline is
resolve -> bag -> build -> threshold; distribution is tree-sharded over
ranks with a forest all-gather (replacing Spark's partition-per-tree
mapPartitions + collect, SharedTrainLogic.scala:129-152, 266-317).
"""

from __future__ import annotations

import logging
from typing import Optional

import torch

from ..core import cpu_engine
from ..core.forest import Forest
from ..utils.params import Params, resolve_params
from ..utils.timing import PhaseTimes, phase
from . import base
from .base import ModelBase, new_uid

logger = logging.getLogger(__name__)


def _shard_range(total: int, rank: int, world: int):
    """Contiguous [lo, hi) split of `total` items over `world` ranks."""
    q, r = divmod(total, world)
    lo = rank * q + min(rank, r)
    return lo, lo + q + (1 if rank < r else 0)


class IsolationForest:
    """Estimator. ``IsolationForest().setNumEstimators(100).fit(X)``.

    Params (names/defaults/validators from
    core/IsolationForestParamsBase.scala:10-109) may also be passed as
    constructor kwargs: ``IsolationForest(numEstimators=200, contamination=0.02)``.
    """

    _params_cls = Params
    _uid_prefix = "isolation-forest"

    def __init__(self, uid: Optional[str] = None, **kwargs):
        self.uid = uid or new_uid(self._uid_prefix)
        self.params = self._params_cls(**kwargs)

    def __getattr__(self, attr):
        # delegate set*/get*/param-name access to the Params registry;
        # setters return the estimator for chaining.
        params = object.__getattribute__(self, "params")
        target = getattr(params, attr)
        if attr.startswith("set") and callable(target):
            def chained(value, _t=target):
                _t(value)
                return self
            return chained
        return target

    # -- fit -----------------------------------------------------------
    def fit(self, data, comm=None) -> "IsolationForestModel":
        """Train. ``data``: [N, d] matrix (numpy / torch / pandas).

        With ``comm`` (a parallel.comm.Comm), ``data`` is this rank's row
        shard; trees are sharded over ranks by GLOBAL tree id and
        all-gathered, so the forest is bit-identical at any world size
        for a fixed seed (when each rank sees the same rows) and
        statistically identical under row sharding.
        """
        times = PhaseTimes()
        X, kind = base.extract_features(data, self.params.get("featuresCol"))
        base.check_output_columns(
            data, kind, self.params.get("scoreCol"), self.params.get("predictionCol")
        )
        seed = self.params.get("randomSeed")
        n_local = X.shape[0]
        total_features = X.shape[1]
        with phase(times, "resolve"):
            if comm is not None:
                total_rows = comm.all_reduce_sum_int(n_local)
            else:
                total_rows = n_local
            rp = resolve_params(self.params, total_rows, total_features)
        if comm is not None and rp.num_samples > n_local:
            raise ValueError(
                f"resolved maxSamples {rp.num_samples} exceeds this rank's local "
                f"row count {n_local}; bags are drawn rank-locally"
            )
        logger.info(
            "fit: %d trees, %d samples/tree, %d/%d features, %d rows",
            self.params.get("numEstimators"), rp.num_samples, rp.num_features,
            total_features, total_rows,
        )

        T = self.params.get("numEstimators")
        rank = comm.rank if comm is not None else 0
        world = comm.world_size if comm is not None else 1
        t_lo, t_hi = _shard_range(T, rank, world)

        with phase(times, "bag+build", device=X if isinstance(X, torch.Tensor) else None):
            forest_local = self._build_local(X, rp, seed, t_lo, t_hi, n_local)
        with phase(times, "gather"):
            forest = self._gather_forest(forest_local, comm, rp, total_features)

        model = self._model_cls()(
            uid=self.uid, forest=forest, params=self.params.copy()
        )
        model._resolved = rp
        if self.params.get("contamination") > 0.0:
            with phase(times, "threshold", device=X if isinstance(X, torch.Tensor) else None):
                scores = model.score(X)
                base.fit_threshold(model, scores, self.params, comm=comm)
        model.fit_metrics = times
        times.log(f"fit[{self.uid}]")
        return model

    # -- internals -------------------------------------------------------
    def _model_cls(self):
        return IsolationForestModel

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

            return gpu_engine.build_forest(
                X, bag_idx, feat_sub, seed, rp, tree_id_offset=t_lo
            )
        if X.is_cuda:
            # GPU build kernel caps bags at 16384 rows (LDS row-index
            # staging); larger bags build on the CPU oracle (bit-identical
            # semantics) and score through the GPU wide-format kernel
            logger.warning(
                "maxSamples %d exceeds the GPU build cap (16384); building "
                "trees on CPU (scoring stays on GPU)", rp.num_samples,
            )
            X = X.float().cpu()
        Xc = X.contiguous().float().numpy()
        return cpu_engine.build_forest(
            Xc, bag_idx, feat_sub, seed, rp.num_samples, rp.num_features,
            rp.total_features, tree_id_offset=t_lo,
        )

    def _gather_forest(self, forest_local, comm, rp, total_features) -> Forest:
        if comm is None:
            return forest_local
        named = dict(
            feature=forest_local.feature,
            value=forest_local.value,
            right=forest_local.right,
            num_instances=forest_local.num_instances,
            node_count=forest_local.node_count,
            value64=forest_local.value64,
        )
        depth_np = getattr(forest_local, "depth_np", None)
        if depth_np is not None:  # GPU builds: keep device packing possible
            named["depth_np"] = depth_np
        arrays = comm.all_gather_forest_arrays(**named)
        depth_np = arrays.pop("depth_np", None)
        forest = Forest(
            num_samples=forest_local.num_samples,
            num_features=forest_local.num_features,
            total_num_features=forest_local.total_num_features,
            **arrays,
        )
        if depth_np is not None:
            forest.depth_np = depth_np
        return forest

    # spark.ml parity helper
    def transformSchema(self, columns):
        """Schema-level validation for column-name workflows
        (Utils.scala:35-65): featuresCol must exist, output cols must not."""
        cols = list(columns)
        if self.params.get("featuresCol") not in cols:
            raise ValueError(f"features column {self.params.get('featuresCol')!r} missing")
        for c in (self.params.get("scoreCol"), self.params.get("predictionCol")):
            if c in cols:
                raise ValueError(f"output column {c!r} already exists")
        return cols + [self.params.get("scoreCol"), self.params.get("predictionCol")]

    @classmethod
    def load(cls, path: str) -> "IsolationForest":
        from ..persist import model_io

        return model_io.load_estimator(cls, path)

    def save(self, path: str, overwrite: bool = False):
        from ..persist import model_io

        model_io.save_estimator(self, path, overwrite=overwrite)
        return self


class IsolationForestModel(ModelBase):
    """Fitted model: an immutable forest + a mutable outlier-score threshold
    (IsolationForestModel.scala:37-96)."""

    _uid_prefix = "isolation-forest"

    def __init__(self, uid: str, forest: Forest, params: Params):
        super().__init__(uid, params)
        self.forest = forest
        self._gpu_forest_cache = {}

    # -- reference-parity accessors -------------------------------------
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
    def isolation_trees(self):
        return self.forest

    def get_num_samples(self) -> int:
        return self.num_samples

    # -- scoring ---------------------------------------------------------
    def _check_scorable(self):
        """IsolationForestModel.scala:118-125: numSamples >= 2 and a
        non-empty forest are required to score."""
        if self.forest.num_trees == 0:
            raise ValueError("this model has no trees; cannot transform")
        if self.forest.num_samples < 2:
            raise ValueError(
                f"numSamples is {self.forest.num_samples}; >= 2 required to transform"
            )

    def score(self, X: torch.Tensor) -> torch.Tensor:
        """Outlier scores, float32 tensor on X's device."""
        self._check_scorable()
        base.validate_feature_vector_size(self.forest.total_num_features, X.shape[1])
        if isinstance(X, torch.Tensor) and X.is_cuda:
            from ..ops import gpu_engine

            return gpu_engine.score_forest(self, X)
        Xc = X.contiguous().float().numpy() if isinstance(X, torch.Tensor) else X
        return torch.from_numpy(cpu_engine.score_forest(self.forest, Xc))

    def transform(self, data, comm=None):
        """Append score + predicted-label columns
        (IsolationForestModel.scala:116-151)."""
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

    # -- persistence ------------------------------------------------------
    def save(self, path: str, overwrite: bool = False):
        from ..persist import model_io

        model_io.save_model(self, path, overwrite=overwrite)
        return self

    @property
    def write(self):
        from ..persist.model_io import WriteHandle

        return WriteHandle(self)

    @classmethod
    def load(cls, path: str) -> "IsolationForestModel":
        from ..persist import model_io

        return model_io.load_model(path, expect_extended=False)
