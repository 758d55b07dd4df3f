"""isolation_forest_amd — an MI355X-native Isolation Forest anomaly-detection engine.

A from-scratch re-design of the capabilities of linkedin/isolation-forest
(Scala/Spark) for AMD Instinct MI355X (gfx950, CDNA4):

* framework layer: PyTorch-ROCm tensors in/out, spark.ml-style
  ``Estimator.fit()`` / ``Model.transform()`` API with the reference's exact
  parameter names, defaults and validators
  (reference: core/IsolationForestParamsBase.scala:10-109),
* hot path: hand-written HIP kernels for bagged tree construction,
  batched path-length scoring and the Extended-IF hyperplane splits
  (reference hot loops: IsolationTree.scala:124-159, 196-230,
  ExtendedIsolationTree.scala:139-260, 283-355),
* scale-out: one process per GPU over RCCL/xGMI via ``torch.distributed``
  (replacing the reference's Spark shuffle/broadcast/collect,
  SharedTrainLogic.scala:129-152, 266-317),
* persistence: the reference's exact on-disk Avro + metadata-JSON model
  format (IsolationForestModelReadWrite.scala:210-323) and an ONNX exporter
  (isolation-forest-onnx/src/isolationforestonnx/isolation_forest_converter.py).

This is NOT a port: no Spark, no JVM, no CUDA shims — the engine is
designed for 288 GB HBM3E per GPU, 64-wide wavefronts and LDS-resident
tree bags.
"""

__version__ = "0.1.0"

from .models.isolation_forest import IsolationForest, IsolationForestModel
from .models.extended_isolation_forest import (
    ExtendedIsolationForest,
    ExtendedIsolationForestModel,
)

__all__ = [
    "IsolationForest",
    "IsolationForestModel",
    "ExtendedIsolationForest",
    "ExtendedIsolationForestModel",
]
