"""Isolation Forest -> ONNX converter (standard IF only).

Produces the same graph shape as the reference's Python converter
(isolation-forest-onnx/src/isolationforestonnx/isolation_forest_converter.py:
120-341): a single ai.onnx.ml TreeEnsembleRegressor with aggregate AVERAGE
whose leaf target weight is depth + c(numInstances), followed by
Div(c(n)) -> Neg -> Pow(2, .) = `outlier_score` and
Less/Not/Cast(threshold) = `predicted_label` (int32). Opsets ai.onnx.ml v1 +
core 14, IR pinned to 10; input `features: [None, numFeatures] float32`.

EIF is NOT exportable (hyperplane splits do not fit the axis-aligned
TreeEnsemble op) — same restriction as the reference (README.md:330-332).

Two entry points:
  IsolationForestConverter(model_file, metadata_file)  — file contract of
      the reference converter (our Avro + metadata JSON are format-
      compatible with the reference's, persist/model_io.py)
  IsolationForestConverter.from_model(model)           — in-memory model
"""

from __future__ import annotations

import json
import math
from typing import Dict

import numpy as np

from ..persist import avro_io
from . import model_proto as mp


def _avg_path_len64(num_instances: int) -> float:
    """Double-precision c(n) — the converter-side formula (the engine's own
    scoring uses the float32 version from utils.math, Utils.scala:85-92)."""
    if num_instances <= 1:
        return 0.0
    n = float(num_instances)
    return 2.0 * (math.log(n - 1.0) + 0.5772156649015329) - (2.0 * (n - 1.0) / n)


def _tree_ensemble_attrs() -> Dict[str, list]:
    return {
        "nodes_treeids": [],
        "nodes_nodeids": [],
        "nodes_featureids": [],
        "nodes_modes": [],
        "nodes_values": [],
        "nodes_truenodeids": [],
        "nodes_falsenodeids": [],
        "nodes_missing_value_tracks_true": [],
        "nodes_hitrates": [],
        "target_treeids": [],
        "target_nodeids": [],
        "target_ids": [],
        "target_weights": [],
    }


class IsolationForestConverter:
    """Converts a saved (or in-memory) standard Isolation Forest model to an
    ONNX model, emitted by the dependency-free writer in model_proto.py."""

    def __init__(self, model_file_path: str, metadata_file_path: str):
        with open(metadata_file_path, "r") as f:
            metadata = json.load(f)
        params = metadata["paramMap"]
        self._num_trees = int(params["numEstimators"])
        self._outlier_score_threshold = float(metadata["outlierScoreThreshold"])
        self._num_samples = int(metadata["numSamples"])
        self._num_features = int(metadata["numFeatures"])
        _, records = avro_io.read_container(model_file_path)
        for r in records:
            if "nodeData" not in r:
                raise ValueError(
                    "extended isolation forest models are not ONNX-exportable"
                )
        self._forest = sorted(records, key=lambda r: int(r["treeID"]))

    @classmethod
    def from_model(cls, model) -> "IsolationForestConverter":
        """Build a converter from an in-memory IsolationForestModel without
        touching disk."""
        from ..models.extended_isolation_forest import ExtendedIsolationForestModel
        from ..persist.model_io import standard_node_records

        if isinstance(model, ExtendedIsolationForestModel):
            raise ValueError(
                "extended isolation forest models are not ONNX-exportable"
            )
        self = cls.__new__(cls)
        forest = model.forest
        self._num_trees = forest.num_trees
        thr = model.outlier_score_threshold
        self._outlier_score_threshold = float(thr if thr is not None else -1.0)
        self._num_samples = forest.num_samples
        self._num_features = (
            forest.total_num_features
            if forest.total_num_features > 0
            else forest.num_features
        )
        self._forest = list(standard_node_records(forest))
        return self

    # ------------------------------------------------------------------

    def convert(self) -> mp.Model:
        attrs = self._build_tree_attrs()
        tree_node = mp.Node(
            op_type="TreeEnsembleRegressor",
            inputs=["features"],
            outputs=["expected_path_len"],
            name="TreeEnsembleRegressorNode",
            domain="ai.onnx.ml",
            attributes=[
                mp.attr_string("aggregate_function", "AVERAGE"),
                mp.attr_int("n_targets", 1),
                mp.attr_ints("nodes_falsenodeids", attrs["nodes_falsenodeids"]),
                mp.attr_ints("nodes_featureids", attrs["nodes_featureids"]),
                mp.attr_floats("nodes_hitrates", attrs["nodes_hitrates"]),
                mp.attr_ints("nodes_missing_value_tracks_true",
                             attrs["nodes_missing_value_tracks_true"]),
                mp.attr_strings("nodes_modes", attrs["nodes_modes"]),
                mp.attr_ints("nodes_nodeids", attrs["nodes_nodeids"]),
                mp.attr_ints("nodes_treeids", attrs["nodes_treeids"]),
                mp.attr_ints("nodes_truenodeids", attrs["nodes_truenodeids"]),
                mp.attr_floats("nodes_values", attrs["nodes_values"]),
                mp.attr_string("post_transform", "NONE"),
                mp.attr_ints("target_ids", attrs["target_ids"]),
                mp.attr_ints("target_nodeids", attrs["target_nodeids"]),
                mp.attr_ints("target_treeids", attrs["target_treeids"]),
                mp.attr_floats("target_weights", attrs["target_weights"]),
            ],
        )
        nodes = [
            tree_node,
            mp.Node("Constant", [], ["avg_path_len"], attributes=[
                mp.attr_tensor("value", mp.Tensor(
                    name="avg_path_len_tensor", data_type=mp.FLOAT, dims=[],
                    float_data=[_avg_path_len64(self._num_samples)]))]),
            mp.Node("Div", ["expected_path_len", "avg_path_len"],
                    ["normalized_path_len"], name="PathLenNormalizer"),
            mp.Node("Neg", ["normalized_path_len"],
                    ["neg_normalized_path_len"], name="NormalizedPathLenNeg"),
            mp.Node("Constant", [], ["constant_2"], attributes=[
                mp.attr_tensor("value", mp.Tensor(
                    name="constant_2", data_type=mp.FLOAT, dims=[],
                    float_data=[2.0]))]),
            mp.Node("Constant", [], ["constant_outlier_score_threshold"],
                    attributes=[
                mp.attr_tensor("value", mp.Tensor(
                    name="constant_outlier_score_threshold",
                    data_type=mp.FLOAT, dims=[],
                    float_data=[self._outlier_score_threshold]))]),
            mp.Node("Pow", ["constant_2", "neg_normalized_path_len"],
                    ["outlier_score"], name="ScoreNode"),
            mp.Node("Less",
                    ["outlier_score", "constant_outlier_score_threshold"],
                    ["less"], name="ScoreLessThanOutlierScoreThreshold"),
            mp.Node("Not", ["less"], ["not"], name="IsOutlier"),
            mp.Node("Cast", ["not"], ["predicted_label"],
                    name="PredictedLabel",
                    attributes=[mp.attr_int("to", mp.INT32)]),
        ]
        graph = mp.Graph(
            name="IsolationForestGraph",
            nodes=nodes,
            inputs=[mp.ValueInfo("features", mp.FLOAT,
                                 [None, self._num_features])],
            outputs=[mp.ValueInfo("outlier_score", mp.FLOAT, [None, 1]),
                     mp.ValueInfo("predicted_label", mp.INT32, [None, 1])],
        )
        return mp.Model(
            graph=graph,
            ir_version=10,
            producer_name="isolation-forest-amd-onnx-converter",
            opset_imports=[mp.OperatorSetId("ai.onnx.ml", 1),
                           mp.OperatorSetId("", 14)],
        )

    def convert_and_save(self, onnx_model_path: str) -> None:
        self.convert().save(onnx_model_path)

    # ------------------------------------------------------------------

    def _build_tree_attrs(self) -> Dict[str, list]:
        attrs = _tree_ensemble_attrs()
        idx = 0
        for tree_id in range(self._num_trees):
            parent_ids: Dict[int, int] = {}
            while (idx < len(self._forest)
                   and int(self._forest[idx]["treeID"]) == tree_id):
                self._add_node_attrs(tree_id, parent_ids,
                                     self._forest[idx]["nodeData"], attrs)
                idx += 1
        return attrs

    @staticmethod
    def _depth_of(parent_ids: Dict[int, int], node_id: int) -> int:
        depth = 0
        while node_id in parent_ids:
            depth += 1
            node_id = parent_ids[node_id]
        return depth

    def _add_node_attrs(self, tree_id: int, parent_ids: Dict[int, int],
                        nd: Dict, attrs: Dict[str, list]) -> None:
        node_id = int(nd["id"])
        left, right = int(nd["leftChild"]), int(nd["rightChild"])
        parent_ids[left] = node_id
        parent_ids[right] = node_id
        is_leaf = left == -1 and right == -1
        attrs["nodes_treeids"].append(tree_id)
        attrs["nodes_nodeids"].append(node_id)
        attrs["nodes_featureids"].append(int(nd["splitAttribute"]))
        attrs["nodes_modes"].append("LEAF" if is_leaf else "BRANCH_LT")
        # nodes_values is an f32 FLOATS attribute but the persisted
        # splitValue is a double; BRANCH_LT tests x < value, and for f32 x
        # and f64 s: x < s <=> x < ceil32(s) (smallest f32 >= s). Emitting
        # the ceil-rounded value makes ONNX inference reproduce the
        # engine's (and Spark's) f64 compare EXACTLY — round-to-nearest
        # (what the reference converter implicitly does) flips knife-edge
        # rows (PARITY.md "Foreign-model f64 split semantics").
        s64 = float(nd["splitValue"])
        s32 = np.float32(s64)
        if not is_leaf and float(s32) < s64:
            s32 = np.nextafter(s32, np.float32(np.inf))
        attrs["nodes_values"].append(float(s32) if not is_leaf else s64)
        attrs["nodes_truenodeids"].append(left)
        attrs["nodes_falsenodeids"].append(right)
        attrs["nodes_missing_value_tracks_true"].append(0)
        attrs["nodes_hitrates"].append(1.0)
        if is_leaf:
            path_len = (self._depth_of(parent_ids, node_id)
                        + _avg_path_len64(int(nd["numInstances"])))
            attrs["target_treeids"].append(tree_id)
            attrs["target_nodeids"].append(node_id)
            attrs["target_ids"].append(0)
            attrs["target_weights"].append(path_len)
