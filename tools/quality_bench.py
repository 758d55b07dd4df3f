#!/usr/bin/env python3
"""Quality benchmark against the reference's published AUROC table.

Reproduces the reference's protocol (README.md:408-415 / BASELINE.md):
100 trees, 256 samples per tree, N trials with unique seeds, mean +/- SEM
of AUROC, on the ODDS datasets bundled with the reference
(mammography 11183x6, shuttle 49097x9). Runs Standard IF, EIF ext=0 and
EIF fully-extended, on CPU or GPU.

    python tools/quality_bench.py [--trials 10] [--device cpu|cuda:0]

Datasets load from the committed in-repo fixtures (tests/fixtures/*.npz)
so this runs on GPU boxes too; prints a markdown table for profiles/.
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from isolation_forest_amd import ExtendedIsolationForest, IsolationForest

FIXTURES = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tests", "fixtures")

PUBLISHED = {  # BASELINE.md (README.md:441-443, 453-455)
    ("mammography", "StandardIF"): (0.8649, 0.0015),
    ("mammography", "ExtendedIF_0"): (0.8633, 0.0007),
    ("mammography", "ExtendedIF_max"): (0.8630, 0.0010),
    ("shuttle", "StandardIF"): (0.9971, 0.0002),
    ("shuttle", "ExtendedIF_0"): (0.9948, 0.0002),
    ("shuttle", "ExtendedIF_max"): (0.9934, 0.0002),
}


def auroc(y, s):
    order = np.argsort(s, kind="mergesort")
    ranks = np.empty(len(s))
    ranks[order] = np.arange(1, len(s) + 1)
    n_pos = int(y.sum())
    n_neg = len(y) - n_pos
    return (ranks[y > 0].sum() - n_pos * (n_pos + 1) / 2) / (n_pos * n_neg)


def load(name):
    with np.load(os.path.join(FIXTURES, f"{name}.npz")) as z:
        return z["X"].astype(np.float32), z["y"].astype(np.float64)


def run(name, X, y, trials, device):
    Xt = torch.from_numpy(X).to(device)
    out = {}
    for label, make in [
        ("StandardIF", lambda s: IsolationForest(
            numEstimators=100, maxSamples=256.0, randomSeed=s)),
        ("ExtendedIF_0", lambda s: ExtendedIsolationForest(
            numEstimators=100, maxSamples=256.0, randomSeed=s,
            extensionLevel=0)),
        ("ExtendedIF_max", lambda s: ExtendedIsolationForest(
            numEstimators=100, maxSamples=256.0, randomSeed=s,
            extensionLevel=X.shape[1] - 1)),
    ]:
        scores = []
        for s in range(1, trials + 1):
            model = make(s).fit(Xt)
            sc = model.score(Xt).float().cpu().numpy()
            scores.append(auroc(y, sc))
        arr = np.array(scores)
        mean, sem = arr.mean(), arr.std(ddof=1) / np.sqrt(len(arr))
        pub = PUBLISHED.get((name, label))
        out[label] = (mean, sem, pub)
        ref = f"{pub[0]:.4f} ± {pub[1]:.4f}" if pub else "—"
        print(f"| {name} | {label} | {mean:.4f} ± {sem:.4f} | {ref} |")
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=10)
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--datasets", nargs="+",
                    default=["mammography", "shuttle"])
    args = ap.parse_args()
    if not os.path.isdir(FIXTURES):
        print("fixtures not found; skipping")
        return
    print("| dataset | model | this engine (mean ± SEM) | reference published |")
    print("|---|---|---|---|")
    for name in args.datasets:
        X, y = load(name)
        run(name, X, y, args.trials, args.device)


if __name__ == "__main__":
    main()
