"""HTTP serving layer tests (FastAPI TestClient, CPU).

Covers: health/model-info endpoints, batch scoring parity with the
engine, label emission under a persisted threshold, dimension/shape
validation, extended-model auto-detection, and serving a reference
Spark-written golden model.
"""

import json
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


class TestServingProcess:
    def test_uvicorn_process_end_to_end(self, saved_model):
        """Real server process (python -m isolation_forest_amd.serving),
        real TCP socket — the deployment path, not just the ASGI shim."""
        import os
        import socket
        import subprocess
        import sys
        import time
        import urllib.request

        _, X, path = saved_model
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        proc = subprocess.Popen(
            [sys.executable, "-m", "isolation_forest_amd.serving",
             "--model", path, "--device", "cpu", "--port", str(port)],
            cwd=repo, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        )
        try:
            deadline = time.time() + 60
            while True:
                try:
                    with urllib.request.urlopen(
                            f"http://127.0.0.1:{port}/healthz", timeout=2) as r:
                        assert r.status == 200
                        break
                except Exception:
                    if time.time() > deadline or proc.poll() is not None:
                        out = proc.stdout.read().decode(errors="replace")
                        raise AssertionError(f"server failed to start: {out[-2000:]}")
                    time.sleep(0.25)
            body = json.dumps({"instances": X[:4].tolist()}).encode()
            req = urllib.request.Request(
                f"http://127.0.0.1:{port}/v1/score", data=body,
                headers={"Content-Type": "application/json"})
            with urllib.request.urlopen(req, timeout=10) as r:
                resp = json.loads(r.read())
            assert len(resp["scores"]) == 4
            assert all(0.0 < s < 1.0 for s in resp["scores"])
        finally:
            proc.terminate()
            proc.wait(timeout=20)
