"""HTTP model serving (FastAPI + uvicorn).

The reference ships batch scoring only (Spark `transform`); production
deployment of anomaly models also needs online scoring, so the MI355X
build provides a serving layer over the same persisted model format:

    python -m isolation_forest_amd.serving --model /path/to/saved_model \
        [--device cuda:0] [--host 0.0.0.0] [--port 8080]

Endpoints (app.py): GET /healthz, GET /v1/model, POST /v1/score. Models
may be engine-written or reference Spark-written directories (standard or
extended — detected from the metadata class). Scoring runs through the
same engine paths as batch transform (HIP kernels on GPU devices).
"""

from .app import create_app, load_any_model

__all__ = ["create_app", "load_any_model"]
