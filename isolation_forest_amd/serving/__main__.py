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
    ap.add_argument("--workers", type=int, default=1)
    args = ap.parse_args()

    import uvicorn

    from .app import create_app

    uvicorn.run(create_app(args.model, args.device),
                host=args.host, port=args.port, workers=args.workers)


if __name__ == "__main__":
    main()
