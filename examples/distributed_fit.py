#!/usr/bin/env python3
"""Multi-GPU (or multi-process CPU) tree-sharded training over RCCL/xGMI.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/distributed_fit.py
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from isolation_forest_amd import IsolationForest
from isolation_forest_amd.parallel import init_from_env


def main():
    comm = init_from_env()  # nccl (=RCCL) on GPU, gloo on CPU
    device = comm.device
    g = torch.Generator(device=device).manual_seed(1234 + comm.rank)
    X_local = torch.randn((2_000_000 if device.type == "cuda" else 20_000, 32),
                          device=device, generator=g)
    if device.type == "cuda":
        X_local = X_local.to(torch.bfloat16)

    # trees sharded over ranks by global tree id, forest all-gathered:
    # every rank ends with the identical full model
    model = IsolationForest(numEstimators=1024, randomSeed=3).fit(
        X_local, comm=comm)
    scores = model.score(X_local)  # no communication: rank-local rows
    mean = scores.float().mean().to(comm.device)
    comm.all_reduce(mean)
    if comm.rank == 0:
        print(f"world={comm.world_size} trees={model.forest.num_trees} "
              f"mean score={float(mean) / comm.world_size:.4f}")


if __name__ == "__main__":
    main()
