#!/usr/bin/env python3
"""Stability soak: repeated fit+score cycles, asserting flat GPU memory.

    python tools/soak.py [--cycles 30] [--rows 2000000]

Covers the production-serving concern the reference delegates to Spark's
executor lifecycle: every cycle builds a fresh model (standard, EIF
fully-extended, EIF ext=0 round-robin), scores, saves+loads every 5th
cycle, and checks torch.cuda.memory_allocated returns to a stable
baseline (no growth across cycles => no tensor/cache leaks).
"""
import argparse
import json
import os
import sys
import tempfile
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cycles", type=int, default=30)
    ap.add_argument("--rows", type=int, default=2_000_000)
    ap.add_argument("--features", type=int, default=16)
    args = ap.parse_args()

    from isolation_forest_amd import (ExtendedIsolationForest,
                                      IsolationForest, IsolationForestModel)
    from isolation_forest_amd.ops import load_extension

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    if dev != "cpu":
        load_extension()
    g = torch.Generator(device=dev)
    g.manual_seed(11)
    X = torch.randn((args.rows, args.features), device=dev,
                    generator=g).to(torch.bfloat16 if dev != "cpu"
                                    else torch.float32)
    base_mem = torch.cuda.memory_allocated() if dev != "cpu" else 0
    mems, t0 = [], time.perf_counter()
    for c in range(args.cycles):
        kind = c % 3
        if kind == 0:
            est = IsolationForest(numEstimators=200, randomSeed=c + 1,
                                  contamination=0.01,
                                  contaminationError=0.001)
        elif kind == 1:
            est = ExtendedIsolationForest(numEstimators=100, randomSeed=c + 1)
        else:
            est = ExtendedIsolationForest(numEstimators=100,
                                          extensionLevel=0, randomSeed=c + 1)
        model = est.fit(X)
        scores = model.score(X)
        assert float(scores.min()) > 0.0 and float(scores.max()) < 1.0
        if c % 5 == 4:
            with tempfile.TemporaryDirectory() as td:
                model.save(os.path.join(td, "m"))
                loaded = (IsolationForestModel.load(os.path.join(td, "m"))
                          if kind == 0 else None)
                del loaded
        del model, scores
        if dev != "cpu":
            torch.cuda.synchronize()
            mems.append(torch.cuda.memory_allocated())
    elapsed = time.perf_counter() - t0
    result = {
        "probe": "soak", "cycles": args.cycles, "rows": args.rows,
        "device": dev, "elapsed_s": round(elapsed, 2),
        "base_mem_mb": round(base_mem / 1e6, 1),
        "mem_first_mb": round(mems[0] / 1e6, 1) if mems else None,
        "mem_last_mb": round(mems[-1] / 1e6, 1) if mems else None,
        "mem_growth_mb": round((mems[-1] - mems[2]) / 1e6, 2)
        if len(mems) > 3 else 0.0,
    }
    print(json.dumps(result))
    if mems and mems[-1] - mems[2] > 50e6:
        print("SOAK_FAIL: memory growth", file=sys.stderr)
        sys.exit(2)
    print("SOAK_OK")


if __name__ == "__main__":
    main()
