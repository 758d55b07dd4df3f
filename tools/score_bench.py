#!/usr/bin/env python3
"""Scoring-kernel microbench: build a forest once, time scoring passes.

Usage: python tools/score_bench.py [--rows 20000000] [--features 32]
       [--trees 1000] [--dtype bf16] [--reps 3] [--extended]

Prints one JSON line with ms per scoring pass and rows/s.
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=20_000_000)
    ap.add_argument("--features", type=int, default=32)
    ap.add_argument("--trees", type=int, default=1000)
    ap.add_argument("--max-samples", type=int, default=256)
    ap.add_argument("--dtype", choices=["bf16", "fp32"], default="bf16")
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--extended", action="store_true")
    ap.add_argument("--extension-level", type=int, default=None)
    args = ap.parse_args()

    from isolation_forest_amd import ExtendedIsolationForest, IsolationForest
    from isolation_forest_amd.ops import load_extension

    load_extension()
    dev = "cuda:0"
    g = torch.Generator(device=dev)
    g.manual_seed(7)
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    # chunked generation keeps peak memory ~= the final matrix (1B x 32
    # bf16 = 64 GB fits one MI355X with room to spare)
    X = torch.empty((args.rows, args.features), device=dev, dtype=dtype)
    step = max(1, min(args.rows, 64_000_000))
    for lo in range(0, args.rows, step):
        hi = min(args.rows, lo + step)
        X[lo:hi] = torch.randn((hi - lo, args.features), device=dev,
                               generator=g).to(dtype)

    cls = ExtendedIsolationForest if args.extended else IsolationForest
    kw = {}
    if args.extended and args.extension_level is not None:
        kw["extensionLevel"] = args.extension_level
    est = cls(numEstimators=args.trees, maxSamples=float(args.max_samples),
              randomSeed=3, **kw)
    t0 = time.perf_counter()
    model = est.fit(X)
    torch.cuda.synchronize()
    t_fit = time.perf_counter() - t0

    for _ in range(args.warmup):
        model.score(X)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        s = model.score(X)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.reps
    print(json.dumps({
        "kernel": "score_extended" if args.extended else "score",
        "ext_level": args.extension_level if args.extended else None,
        "rows": args.rows, "features": args.features, "trees": args.trees,
        "dtype": args.dtype, "fit_s": round(t_fit, 4),
        "score_ms": round(dt * 1e3, 3),
        "score_rows_per_s": round(args.rows / dt),
        "sample_scores": [round(float(v), 6) for v in s[:4].float().cpu()],
    }))


if __name__ == "__main__":
    main()
