"""Typed parameter registry with the reference's names/defaults/validators.

Mirrors core/IsolationForestParamsBase.scala:10-109 and
extended/ExtendedIsolationForestParams.scala:19-28:

* numEstimators:      int   > 0,          default 100
* maxSamples:         float > 0,          default 256   (<=1.0 => fraction of rows, >1 => count)
* contamination:      float in [0, 0.5),  default 0.0   (0.0 => threshold not set)
* contaminationError: float in [0, 1],    default 0.0   (0.0 => exact quantile)
* maxFeatures:        float > 0,          default 1.0   (<=1.0 => fraction of features, >1 => count)
* bootstrap:          bool,               default False
* randomSeed:         int   > 0,          default 1
* featuresCol / predictionCol / scoreCol: str, defaults "features"/"predictedLabel"/"outlierScore"
* extensionLevel (EIF only): int >= 0, no default (resolved at fit time to
  numFeatures-1 when unset — ExtendedIsolationForest.scala:57-69)

The paramMap is part of the on-disk model format
(IsolationForestModelReadWriteUtils.scala:163-187) and is re-applied on
load; JSON types here match what Spark's Param.jsonEncode produced in the
committed golden fixtures (maxSamples/maxFeatures as JSON numbers, etc.).
"""

from __future__ import annotations

import dataclasses
from typing import Any, Callable, Dict, Optional


@dataclasses.dataclass
class ParamSpec:
    name: str
    ptype: type
    default: Any
    validator: Callable[[Any], bool]
    doc: str = ""

    def validate(self, value):
        if self.ptype is float and isinstance(value, int) and not isinstance(value, bool):
            value = float(value)
        if self.ptype is int and isinstance(value, bool):
            raise TypeError(f"{self.name} must be {self.ptype.__name__}, got bool")
        if not isinstance(value, self.ptype):
            raise TypeError(
                f"{self.name} must be {self.ptype.__name__}, got {type(value).__name__}"
            )
        if not self.validator(value):
            raise ValueError(f"{self.name}={value!r} failed validation: {self.doc}")
        return value


BASE_PARAMS = [
    ParamSpec("numEstimators", int, 100, lambda v: v > 0, "number of trees > 0"),
    ParamSpec(
        "maxSamples", float, 256.0, lambda v: v > 0,
        "samples per tree; <=1.0 fraction of rows, >1 count",
    ),
    ParamSpec(
        "contamination", float, 0.0, lambda v: 0.0 <= v < 0.5,
        "expected outlier fraction in [0, 0.5)",
    ),
    ParamSpec(
        "contaminationError", float, 0.0, lambda v: 0.0 <= v <= 1.0,
        "allowed quantile error in [0, 1]; 0.0 = exact",
    ),
    ParamSpec(
        "maxFeatures", float, 1.0, lambda v: v > 0,
        "features per tree; <=1.0 fraction, >1 count",
    ),
    ParamSpec("bootstrap", bool, False, lambda v: True, "sample with replacement"),
    ParamSpec("randomSeed", int, 1, lambda v: v > 0, "random seed > 0"),
    ParamSpec("featuresCol", str, "features", lambda v: len(v) > 0, "input column"),
    ParamSpec(
        "predictionCol", str, "predictedLabel", lambda v: len(v) > 0, "label column"
    ),
    ParamSpec("scoreCol", str, "outlierScore", lambda v: len(v) > 0, "score column"),
]

EXTENDED_PARAMS = BASE_PARAMS + [
    ParamSpec(
        "extensionLevel", int, None, lambda v: v >= 0,
        "extensionLevel+1 = non-zero hyperplane coordinates; 0 = axis-aligned; "
        "unset = fully extended (numFeatures-1) at fit time",
    ),
]


class Params:
    """Param holder with Spark-style setters/getters and dict round-trip."""

    SPECS: Dict[str, ParamSpec] = {p.name: p for p in BASE_PARAMS}

    def __init__(self, **kwargs):
        self._values: Dict[str, Any] = {}
        for name, value in kwargs.items():
            self.set(name, value)

    # -- generic access ------------------------------------------------
    def set(self, name: str, value):
        spec = self.SPECS.get(name)
        if spec is None:
            raise KeyError(f"unknown param {name!r}")
        if value is None:
            self._values.pop(name, None)
            return self
        self._values[name] = spec.validate(value)
        return self

    def get(self, name: str):
        spec = self.SPECS.get(name)
        if spec is None:
            raise KeyError(f"unknown param {name!r}")
        if name in self._values:
            return self._values[name]
        if spec.default is None:
            raise KeyError(f"param {name!r} is not set and has no default")
        return spec.default

    def is_set(self, name: str) -> bool:
        return name in self._values

    def to_dict(self) -> Dict[str, Any]:
        """Full effective map (defaults included) for metadata JSON."""
        out = {}
        for name, spec in self.SPECS.items():
            if name in self._values:
                out[name] = self._values[name]
            elif spec.default is not None:
                out[name] = spec.default
        return out

    def copy(self) -> "Params":
        new = type(self)()
        new._values = dict(self._values)
        return new

    def apply_map(self, mapping: Dict[str, Any]):
        """Re-apply a persisted paramMap (model load path)."""
        for name, value in mapping.items():
            if name in self.SPECS:
                self.set(name, value)
        return self

    # -- Spark-style accessors ----------------------------------------
    def __getattr__(self, attr: str):
        # setNumEstimators / getNumEstimators style
        if attr.startswith("set") and len(attr) > 3:
            pname = attr[3].lower() + attr[4:]
            if pname in self.SPECS:
                def setter(value, _p=pname):
                    self.set(_p, value)
                    return self
                return setter
        if attr.startswith("get") and len(attr) > 3:
            pname = attr[3].lower() + attr[4:]
            if pname in self.SPECS:
                return lambda _p=pname: self.get(_p)
        if attr in self.SPECS:
            return self.get(attr)
        raise AttributeError(attr)


class ExtendedParams(Params):
    SPECS: Dict[str, ParamSpec] = {p.name: p for p in EXTENDED_PARAMS}


@dataclasses.dataclass
class ResolvedParams:
    """Parameters resolved against a concrete dataset.

    Mirrors core/Utils.scala:12-16 + SharedTrainLogic.validateAndResolveParams
    (SharedTrainLogic.scala:27-78): fraction-vs-count resolution for
    maxSamples/maxFeatures, plus dataset facts.
    """

    num_samples: int  # rows per tree (resolved maxSamples)
    num_features: int  # features per tree (resolved maxFeatures)
    total_rows: int
    total_features: int
    extension_level: Optional[int] = None  # EIF only (resolved)


def resolve_params(params: Params, total_rows: int, total_features: int) -> ResolvedParams:
    """Resolve fractions to counts and validate against dataset shape.

    Reference: SharedTrainLogic.scala:27-78 (maxFeatures via numFeatures from
    head(), maxSamples via count(); requires numFeatures > 0 and
    numSamples in [2, totalRows])."""
    if total_features <= 0:
        raise ValueError("dataset must have at least one feature")
    if total_rows < 2:
        raise ValueError("dataset must contain at least 2 rows to fit a model")

    mf = params.get("maxFeatures")
    num_features = int(mf * total_features) if mf <= 1.0 else int(mf)
    if num_features <= 0:
        raise ValueError(
            f"resolved maxFeatures {num_features} (maxFeatures={mf}, "
            f"totalFeatures={total_features}) must be > 0"
        )
    if num_features > total_features:
        raise ValueError(
            f"resolved maxFeatures {num_features} exceeds feature count {total_features}"
        )

    ms = params.get("maxSamples")
    num_samples = int(ms * total_rows) if ms <= 1.0 else int(ms)
    if num_samples < 2:
        raise ValueError(
            f"resolved maxSamples {num_samples} (maxSamples={ms}) must be >= 2"
        )
    if num_samples > total_rows:
        raise ValueError(
            f"resolved maxSamples {num_samples} exceeds row count {total_rows}; "
            "set maxSamples <= number of rows"
        )

    ext = None
    if params.is_set("extensionLevel") if isinstance(params, ExtendedParams) else False:
        ext = params.get("extensionLevel")
    if isinstance(params, ExtendedParams):
        if ext is None:
            ext = num_features - 1  # fully extended (ExtendedIsolationForest.scala:57-63)
        if ext < 0 or ext > num_features - 1:
            raise ValueError(
                f"extensionLevel {ext} must be in [0, {num_features - 1}] for the "
                f"resolved {num_features}-dimensional feature subspace"
            )

    return ResolvedParams(
        num_samples=num_samples,
        num_features=num_features,
        total_rows=total_rows,
        total_features=total_features,
        extension_level=ext,
    )
