"""ONNX exporter tests (CPU): graph parity with the engine at 1e-5, wire
format round-trip, depth/avg-path-length computation, file-based converter
contract, EIF export rejection.

Mirrors the reference's two-part engine<->ONNX integration test
(test_isolation_forest_onnx_integration.py:12-89, tolerance 1e-5) with the
bundled evaluator standing in for onnxruntime (not installed here).
"""

import glob
import json
import math
import os

import numpy as np
import pytest
import torch

from isolation_forest_amd import ExtendedIsolationForest, IsolationForest
from isolation_forest_amd.onnx import IsolationForestConverter, evaluator
from isolation_forest_amd.onnx.converter import _avg_path_len64
from tests.conftest import auroc


def make_data(n=4000, d=6, seed=5, outliers=80):
    rs = np.random.RandomState(seed)
    inl = rs.normal(size=(n, d)).astype(np.float32)
    out = rs.uniform(-8, 9, size=(outliers, d)).astype(np.float32)
    X = np.concatenate([inl, out])
    y = np.concatenate([np.zeros(n), np.ones(outliers)])
    return X, y


@pytest.fixture(scope="module")
def trained():
    X, y = make_data()
    model = IsolationForest(
        numEstimators=100, contamination=0.02, contaminationError=0.01,
        randomSeed=11,
    ).fit(torch.from_numpy(X))
    return X, y, model


class TestScoreParity:
    def test_engine_vs_onnx_1e5(self, trained):
        X, y, model = trained
        onnx_bytes = IsolationForestConverter.from_model(model).convert().serialize()
        res = evaluator.run(onnx_bytes, X)
        onnx_scores = res["outlier_score"].ravel()
        engine_scores = model.score(torch.from_numpy(X)).numpy()
        assert np.abs(onnx_scores - engine_scores).max() < 1e-5

    def test_labels_match_transform(self, trained):
        X, y, model = trained
        onnx_bytes = IsolationForestConverter.from_model(model).convert().serialize()
        res = evaluator.run(onnx_bytes, X)
        labels = res["predicted_label"].ravel()
        expect = model.transform(torch.from_numpy(X))["predictedLabel"].numpy()
        np.testing.assert_array_equal(labels, expect.astype(np.int32))

    def test_auroc_preserved(self, trained):
        X, y, model = trained
        onnx_bytes = IsolationForestConverter.from_model(model).convert().serialize()
        scores = evaluator.run(onnx_bytes, X)["outlier_score"].ravel()
        assert auroc(y, scores) > 0.95


class TestFileContract:
    def test_convert_from_saved_model(self, trained, tmp_path):
        """The reference converter's ctor contract: explicit Avro + metadata
        file paths (isolation_forest_converter.py:54-96)."""
        X, y, model = trained
        path = str(tmp_path / "saved")
        model.save(path)
        avro_file = glob.glob(os.path.join(path, "data", "*.avro"))[0]
        meta_file = glob.glob(os.path.join(path, "metadata", "part-*"))[0]
        conv = IsolationForestConverter(avro_file, meta_file)
        out_path = str(tmp_path / "model.onnx")
        conv.convert_and_save(out_path)
        onnx_bytes = open(out_path, "rb").read()
        scores = evaluator.run(onnx_bytes, X)["outlier_score"].ravel()
        engine_scores = model.score(torch.from_numpy(X)).numpy()
        assert np.abs(scores - engine_scores).max() < 1e-5

    def test_extended_model_rejected(self, tmp_path):
        X, _ = make_data(n=1500, d=4, outliers=0)
        model = ExtendedIsolationForest(numEstimators=10).fit(
            torch.from_numpy(X))
        with pytest.raises(ValueError, match="not ONNX-exportable"):
            IsolationForestConverter.from_model(model)
        path = str(tmp_path / "eif")
        model.save(path)
        avro_file = glob.glob(os.path.join(path, "data", "*.avro"))[0]
        meta_file = glob.glob(os.path.join(path, "metadata", "part-*"))[0]
        with pytest.raises(ValueError, match="not ONNX-exportable"):
            IsolationForestConverter(avro_file, meta_file)


class TestGraphStructure:
    def test_model_header(self, trained):
        _, _, model = trained
        m = IsolationForestConverter.from_model(model).convert()
        b = m.serialize()
        parsed = evaluator.parse_model(b)
        assert parsed["ir_version"] == 10
        assert ("ai.onnx.ml", 1) in parsed["opsets"]
        assert ("", 14) in parsed["opsets"]
        assert parsed["inputs"] == ["features"]
        assert parsed["outputs"] == ["outlier_score", "predicted_label"]
        ops = [n["op_type"] for n in parsed["nodes"]]
        assert ops.count("TreeEnsembleRegressor") == 1
        for op in ["Div", "Neg", "Pow", "Less", "Not", "Cast"]:
            assert op in ops

    def test_tree_attrs(self, trained):
        _, _, model = trained
        parsed = evaluator.parse_model(
            IsolationForestConverter.from_model(model).convert().serialize())
        tree = next(n for n in parsed["nodes"]
                    if n["op_type"] == "TreeEnsembleRegressor")
        a = tree["attrs"]
        assert a["aggregate_function"] == b"AVERAGE"
        assert a["post_transform"] == b"NONE"
        assert a["n_targets"] == 1
        n_nodes = len(a["nodes_nodeids"])
        for k in ["nodes_treeids", "nodes_featureids", "nodes_modes",
                  "nodes_values", "nodes_truenodeids", "nodes_falsenodeids",
                  "nodes_hitrates", "nodes_missing_value_tracks_true"]:
            assert len(a[k]) == n_nodes, k
        n_leaves = sum(1 for m in a["nodes_modes"] if m == "LEAF")
        assert len(a["target_weights"]) == n_leaves
        assert set(a["nodes_modes"]) == {"LEAF", "BRANCH_LT"}
        # leaves carry sentinel children and feature (reference converter
        # passes the Avro sentinels through)
        for i, m in enumerate(a["nodes_modes"]):
            if m == "LEAF":
                assert a["nodes_truenodeids"][i] == -1
                assert a["nodes_falsenodeids"][i] == -1
                assert a["nodes_featureids"][i] == -1

    def test_leaf_weights_are_depth_plus_c(self, trained):
        _, _, model = trained
        forest = model.forest
        parsed = evaluator.parse_model(
            IsolationForestConverter.from_model(model).convert().serialize())
        tree = next(n for n in parsed["nodes"]
                    if n["op_type"] == "TreeEnsembleRegressor")
        a = tree["attrs"]
        from isolation_forest_amd.ops.gpu_engine import _node_depths

        depth = _node_depths(forest.feature, forest.right)
        w = {(t, n): wt for t, n, wt in zip(
            a["target_treeids"], a["target_nodeids"], a["target_weights"])}
        for t in range(forest.num_trees):
            for i in range(int(forest.node_count[t])):
                if forest.feature[t, i] == -1:
                    expect = np.float32(
                        depth[t, i]
                        + _avg_path_len64(int(forest.num_instances[t, i])))
                    assert w[(t, i)] == pytest.approx(float(expect), abs=2e-6)


class TestAvgPathLen:
    """Goldens from the reference converter tests
    (test_isolation_forest_converter.py:162-174)."""

    def test_small_values(self):
        assert _avg_path_len64(0) == 0.0
        assert _avg_path_len64(1) == 0.0
        assert _avg_path_len64(2) == pytest.approx(
            2 * (math.log(1) + np.euler_gamma) - 1.0)

    def test_formula(self):
        for n in [3, 10, 256, 4096]:
            expect = 2 * (np.log(n - 1) + np.euler_gamma) - 2 * (n - 1) / n
            assert _avg_path_len64(n) == pytest.approx(expect, rel=1e-12)


class TestWireFormat:
    def test_roundtrip_scalars(self):
        from isolation_forest_amd.onnx import protowire as pw

        for v in [0, 1, 127, 128, 300, 2 ** 40, -1, -300]:
            enc = pw.varint(v)
            dec, pos = pw.read_varint(enc, 0)
            assert pos == len(enc)
            assert pw.as_int64(dec) == v

    def test_metadata_json_contract(self, trained, tmp_path):
        _, _, model = trained
        path = str(tmp_path / "m")
        model.save(path)
        meta = json.loads(
            open(glob.glob(os.path.join(path, "metadata", "part-*"))[0]).read())
        # fields the reference converter requires (:57-74)
        assert "outlierScoreThreshold" in meta
        assert "numSamples" in meta and "numFeatures" in meta
        for k in ["numEstimators", "maxFeatures", "contamination",
                  "maxSamples"]:
            assert k in meta["paramMap"]


from tests.conftest import GOLDEN


class TestReferenceModelConversion:
    """Convert the REFERENCE's golden Spark-written model through our
    converter (the reference converter's exact file contract) and check
    score parity against our engine loading the same model — the analog of
    the reference's two-part Spark<->ONNX integration test at 1e-5."""

    def test_reference_golden_model_onnx_parity(self, mammography):
        import glob as _glob

        from isolation_forest_amd import IsolationForestModel

        mdir = os.path.join(GOLDEN, "savedIsolationForestModel")
        avro_file = _glob.glob(os.path.join(mdir, "data", "*.avro"))[0]
        meta_file = _glob.glob(os.path.join(mdir, "metadata", "part-*"))[0]
        conv = IsolationForestConverter(avro_file, meta_file)
        onnx_bytes = conv.convert().serialize()

        model = IsolationForestModel.load(mdir)
        X = mammography[0][:2000]
        engine_scores = model.score(torch.from_numpy(X)).numpy()
        onnx_scores = evaluator.run(onnx_bytes, X)["outlier_score"].ravel()
        assert np.abs(onnx_scores - engine_scores).max() < 1e-5


class TestUnsetThresholdExport:
    def test_labels_match_reference_converter_semantics(self, gaussian_data):
        """Threshold sentinel -1: the exported graph's Less/Not chain makes
        every label 1 (score >= -1) — byte-for-byte what the REFERENCE
        converter emits for such metadata (its graph is identical); the
        engines' native transform instead emits literal 0.0 labels
        (IsolationForestModel.scala:143-148). Pinned so the divergence is
        intentional, not accidental."""
        X, _ = gaussian_data
        from isolation_forest_amd import IsolationForest

        m = IsolationForest(numEstimators=10, randomSeed=2).fit(X[:800])
        assert m.outlier_score_threshold == -1.0
        data = IsolationForestConverter.from_model(m).convert().serialize()
        r = evaluator.run(data, X[:16])
        assert (r["predicted_label"].ravel() == 1).all()
        native = m.transform(X[:16])["predictedLabel"]
        assert float(native.sum()) == 0.0
