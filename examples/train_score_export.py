#!/usr/bin/env python3
"""Single-GPU end-to-end example: train, threshold, save, reload, ONNX.

    python examples/train_score_export.py [--device cuda:0] [--rows 1000000]
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from isolation_forest_amd import IsolationForest, IsolationForestModel
from isolation_forest_amd.onnx import IsolationForestConverter, evaluator


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    ap.add_argument("--rows", type=int, default=None,
                    help="default: 1M on GPU, 100k on CPU")
    ap.add_argument("--features", type=int, default=32)
    ap.add_argument("--out", default="/tmp/ifa_example")
    args = ap.parse_args()
    if args.rows is None:
        args.rows = 1_000_000 if args.device.startswith("cuda") else 100_000

    g = torch.Generator(device=args.device).manual_seed(0)
    X = torch.randn((args.rows, args.features), device=args.device, generator=g)
    if args.device.startswith("cuda"):
        X = X.to(torch.bfloat16)

    model = (IsolationForest()
             .setNumEstimators(200)
             .setContamination(0.01)
             .setContaminationError(0.001)
             .setRandomSeed(7)
             .fit(X))
    out = model.transform(X)
    print("flagged:", float(out["predictedLabel"].float().mean()))
    print("fit phases:", dict(model.fit_metrics))

    model.save(args.out, overwrite=True)
    reloaded = IsolationForestModel.load(args.out)
    print("reload OK; threshold:", reloaded.outlier_score_threshold)

    onnx_path = args.out + ".onnx"
    IsolationForestConverter.from_model(model).convert_and_save(onnx_path)
    sample = X[:1000].float().cpu().numpy()
    res = evaluator.run(open(onnx_path, "rb").read(), sample)
    engine = model.score(X[:1000]).float().cpu().numpy()
    print("onnx max |diff|:", float(np.abs(res["outlier_score"].ravel() - engine).max()))


if __name__ == "__main__":
    main()
