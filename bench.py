#!/usr/bin/env python3
"""Flagship benchmark: train+score rows/sec for Isolation Forest on MI355X.

Driver contract:
    python bench.py --gpus N --steps K --warmup W
(N>1 is launched via torch.distributed.run, one rank per GPU over RCCL.)

One "step" = one full training pass over the fixed synthetic matrix
(resolve -> bag -> GPU tree build -> forest all-gather) PLUS scoring every
row (the batched path-length traversal kernel). The headline config is
BASELINE.json's: Standard IF, 1000 trees, 256-sample bags, 100M rows x 32
features bf16 per GPU (weak scaling), synthetic Gaussian-mixture anomalies
with random seeds. Rank 0 prints ONE JSON line.
"""

import argparse
import json
import os
import socket
import sys
import time

import numpy as np
import torch


def _free_port() -> int:
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _self_launch(n: int) -> int:
    """Re-exec this script under torch.distributed.run with n ranks.

    The driver may call ``python bench.py --gpus N`` directly; the contract
    requires one rank per GPU over RCCL, so when WORLD_SIZE is unset we
    self-launch through the elastic launcher (127.0.0.1 rendezvous — the
    container hostname may not resolve).
    """
    import subprocess

    cmd = [
        sys.executable,
        "-m",
        "torch.distributed.run",
        "--nnodes=1",
        f"--nproc-per-node={n}",
        "--master-addr=127.0.0.1",
        f"--master-port={_free_port()}",
        os.path.abspath(__file__),
    ] + sys.argv[1:]
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    return subprocess.call(cmd, env=env)


def make_data(rows, d, device, dtype, seed, outlier_frac=0.005):
    """Synthetic Gaussian-mixture anomaly data + labels (on device)."""
    g = torch.Generator(device=device)
    g.manual_seed(seed)
    n_out = int(rows * outlier_frac)
    n_in = rows - n_out
    X = torch.empty((rows, d), device=device, dtype=torch.float32)
    # inliers: two tight clusters; outliers: broad uniform background
    half = n_in // 2
    X[:half].normal_(0.0, 1.0, generator=g)
    X[half:n_in].normal_(4.0, 0.7, generator=g)
    X[n_in:].uniform_(-12.0, 16.0, generator=g)
    y = torch.zeros(rows, device=device, dtype=torch.bool)
    y[n_in:] = True
    perm = torch.randperm(rows, device=device, generator=g)
    X = X[perm].contiguous()
    y = y[perm]
    if dtype == "bf16":
        X = X.to(torch.bfloat16)
    return X, y


def auroc_torch(y: torch.Tensor, scores: torch.Tensor) -> float:
    scores = scores.float()
    order = torch.argsort(scores)
    ranks = torch.empty_like(scores)
    ranks[order] = torch.arange(
        1, scores.numel() + 1, device=scores.device, dtype=torch.float32
    )
    pos = y
    n_pos = int(pos.sum())
    n_neg = scores.numel() - n_pos
    if n_pos == 0 or n_neg == 0:
        return float("nan")
    return float(
        (ranks[pos].sum() - n_pos * (n_pos + 1) / 2.0) / (float(n_pos) * n_neg)
    )


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--rows", type=int, default=None, help="rows PER GPU")
    ap.add_argument("--features", type=int, default=32)
    ap.add_argument("--trees", type=int, default=1000)
    ap.add_argument("--max-samples", type=int, default=256)
    ap.add_argument("--extended", action="store_true")
    ap.add_argument("--dtype", choices=["bf16", "fp32"], default="bf16")
    ap.add_argument("--contamination", type=float, default=0.0)
    ap.add_argument("--device", default=None)
    ap.add_argument("--export-onnx", default=None, metavar="PATH",
                    help="after timing, rank 0 exports the trained model's "
                         "ONNX graph here (config #5's export leg; "
                         "standard IF only)")
    args = ap.parse_args()

    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        sys.exit(_self_launch(args.gpus))

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from isolation_forest_amd import ExtendedIsolationForest, IsolationForest

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    have_gpu = torch.cuda.is_available()
    device = args.device or ("cuda" if have_gpu else "cpu")
    if device == "cpu":
        # plumbing mode for GPU-less containers: tiny config, same code path
        rows = args.rows or 20000
        trees = min(args.trees, 100)
        dtype = "fp32"
    else:
        from isolation_forest_amd.ops import load_extension

        load_extension()  # HIP extension is mandatory on the GPU path
        rows = args.rows or 100_000_000
        trees = args.trees
        dtype = args.dtype
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"

    comm = None
    if world_size > 1:
        from isolation_forest_amd.parallel import init_from_env

        comm = init_from_env()

    X, y = make_data(rows, args.features, device, dtype, seed=1234 + rank * 7919)

    est_cls = ExtendedIsolationForest if args.extended else IsolationForest
    model_name = "ExtendedIsolationForest" if args.extended else "IsolationForest"

    def one_step(step_idx: int):
        est = est_cls(
            numEstimators=trees,
            maxSamples=float(args.max_samples),
            contamination=args.contamination,
            contaminationError=0.01 if args.contamination > 0 else 0.0,
            randomSeed=1 + step_idx,
        )
        model = est.fit(X, comm=comm)
        scores = model.score(X)
        return model, scores

    def sync():
        if device.startswith("cuda"):
            torch.cuda.synchronize()
        if comm is not None:
            comm.barrier()

    for w in range(args.warmup):
        one_step(1000 + w)
    sync()

    t0 = time.perf_counter()
    model, scores = None, None
    for s in range(args.steps):
        model, scores = one_step(s)
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if comm is not None:
        elapsed = comm.all_reduce_max(elapsed)  # MAX over ranks
        comm.barrier()

    rows_total = rows * world_size
    ms_per_step = elapsed / args.steps * 1000.0
    rows_per_sec = rows_total * args.steps / elapsed

    # quality (outside the timed region): AUROC on this rank's shard
    sample = min(rows, 10_000_000)
    measured_auroc = auroc_torch(y[:sample], scores[:sample])

    if rank == 0 and args.export_onnx and not args.extended:
        from isolation_forest_amd.onnx import IsolationForestConverter

        IsolationForestConverter.from_model(model).convert_and_save(
            args.export_onnx)

    if rank == 0:
        result = {
            "metric": "train+score rows/sec (whole node) + AUROC, 100Mx32 IF",
            "value": rows_per_sec,
            "unit": "rows/sec",
            "n_gpus": world_size if device.startswith("cuda") else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": dtype,
            "data": "synthetic gaussian-mixture anomalies (0.5% planted outliers), random seeds",
            "config": {
                "model": model_name,
                "trees": trees,
                "max_samples": args.max_samples,
                "rows_per_gpu": rows,
                "rows_total": rows_total,
                "features": args.features,
                "parallelism": f"row-sharded dp{world_size} + tree-sharded build",
                "world_size": world_size,
                "contamination": args.contamination,
                "auroc": round(measured_auroc, 4),
                "device": device.split(":")[0],
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
