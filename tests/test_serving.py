"""HTTP serving layer tests (FastAPI TestClient, CPU).

Covers: health/model-info endpoints, batch scoring parity with the
engine, label emission under a persisted threshold, dimension/shape
validation, extended-model auto-detection, and serving a reference
Spark-written golden model.
"""

import numpy as np
import pytest
import torch
from fastapi.testclient import TestClient

from isolation_forest_amd import ExtendedIsolationForest, IsolationForest
from isolation_forest_amd.serving import create_app
from tests.conftest import GOLDEN


@pytest.fixture(scope="module")
def saved_model(tmp_path_factory, gaussian_data):
    X, _ = gaussian_data
    model = IsolationForest(
        numEstimators=30, contamination=0.05, randomSeed=21
    ).fit(X)
    path = str(tmp_path_factory.mktemp("serve") / "model")
    model.save(path)
    return model, X, path


class TestServing:
    def test_health_and_info(self, saved_model):
        model, _, path = saved_model
        client = TestClient(create_app(path, device="cpu"))
        assert client.get("/healthz").json() == {"status": "ok"}
        info = client.get("/v1/model").json()
        assert info["class"] == "IsolationForestModel"
        assert info["numTrees"] == 30
        assert info["numFeatures"] == model.num_features
        assert info["outlierScoreThreshold"] == pytest.approx(
            model.outlier_score_threshold)

    def test_score_parity_and_labels(self, saved_model):
        model, X, path = saved_model
        client = TestClient(create_app(path, device="cpu"))
        rows = X[:64]
        r = client.post("/v1/score", json={"instances": rows.tolist()})
        assert r.status_code == 200
        body = r.json()
        served = np.asarray(body["scores"], dtype=np.float32)
        engine = model.score(torch.from_numpy(rows)).numpy()
        np.testing.assert_allclose(served, engine, rtol=0, atol=1e-6)
        labels = np.asarray(body["labels"])
        np.testing.assert_array_equal(
            labels, (engine >= model.outlier_score_threshold).astype(int))

    def test_dimension_validation(self, saved_model):
        _, _, path = saved_model
        client = TestClient(create_app(path, device="cpu"))
        r = client.post("/v1/score", json={"instances": [[1.0, 2.0]]})
        assert r.status_code == 400
        assert "features" in r.json()["detail"]
        r = client.post("/v1/score",
                        json={"instances": [[1, 2, 3], [1, 2]]})
        assert r.status_code == 400
        r = client.post("/v1/score", json={"instances": []})
        assert r.status_code == 200
        assert r.json()["scores"] == []

    def test_extended_autodetect_no_labels_without_threshold(
            self, tmp_path, gaussian_data):
        X, _ = gaussian_data
        model = ExtendedIsolationForest(numEstimators=12, randomSeed=4).fit(X)
        path = str(tmp_path / "ext")
        model.save(path)
        client = TestClient(create_app(path, device="cpu"))
        info = client.get("/v1/model").json()
        assert info["class"] == "ExtendedIsolationForestModel"
        r = client.post("/v1/score", json={"instances": X[:8].tolist()})
        assert r.status_code == 200
        assert r.json()["labels"] is None

    def test_serves_reference_golden_model(self, mammography):
        import os

        path = os.path.join(GOLDEN, "savedIsolationForestModel")
        client = TestClient(create_app(path, device="cpu"))
        info = client.get("/v1/model").json()
        assert info["numTrees"] == 100
        X, _ = mammography
        r = client.post("/v1/score", json={"instances": X[:16].tolist()})
        assert r.status_code == 200
        scores = r.json()["scores"]
        assert len(scores) == 16
        assert all(0.0 < s < 1.0 for s in scores)
