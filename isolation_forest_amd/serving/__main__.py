"""CLI entry: python -m isolation_forest_amd.serving --model PATH ..."""

import argparse


def main():
    ap = argparse.ArgumentParser(
        prog="isolation_forest_amd.serving",
        description="Serve a persisted Isolation Forest model over HTTP")
    ap.add_argument("--model", required=True, help="saved model directory")
    ap.add_argument("--device", default=None,
                    help="cuda:N or cpu (default: cuda:0 when available)")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8080)
    args = ap.parse_args()

    import uvicorn

    from .app import create_app

    # single worker by design: one model copy per process; scale out with
    # multiple processes behind a load balancer (one per GPU)
    uvicorn.run(create_app(args.model, args.device),
                host=args.host, port=args.port)


if __name__ == "__main__":
    main()
