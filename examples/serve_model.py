#!/usr/bin/env python3
"""Train, persist and serve a model over HTTP, then query it.

    python examples/serve_model.py

Starts the FastAPI service in-process (TestClient — swap for
`python -m isolation_forest_amd.serving --model <dir>` + any HTTP client
in production) and scores a batch through it.
"""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from isolation_forest_amd import IsolationForest  # noqa: E402
from isolation_forest_amd.serving import create_app  # noqa: E402


def main():
    rs = np.random.RandomState(0)
    X = np.concatenate([
        rs.normal(size=(5000, 8)).astype(np.float32),
        rs.uniform(-8, 8, size=(100, 8)).astype(np.float32),
    ])
    model = IsolationForest(
        numEstimators=100, contamination=0.02, randomSeed=1).fit(X)

    with tempfile.TemporaryDirectory() as td:
        path = os.path.join(td, "model")
        model.save(path)

        from fastapi.testclient import TestClient

        client = TestClient(create_app(path))
        print("model info:", client.get("/v1/model").json())
        resp = client.post("/v1/score",
                           json={"instances": X[-5:].tolist()}).json()
        for row, s, lab in zip(X[-5:], resp["scores"], resp["labels"]):
            print(f"score={s:.4f} label={lab}")


if __name__ == "__main__":
    main()
