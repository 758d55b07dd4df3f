"""ONNX export for standard Isolation Forest models (converter.py), with a
dependency-free protobuf writer (model_proto.py / protowire.py) and a
numpy evaluator of the exported graph (evaluator.py) for environments
without onnxruntime. Mirrors the reference's isolation-forest-onnx package
(isolation_forest_converter.py)."""

from .converter import IsolationForestConverter
from . import evaluator

__all__ = ["IsolationForestConverter", "evaluator"]
