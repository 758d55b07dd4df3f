#!/usr/bin/env python3
"""Two-rank RCCL evidence probe.

Launched via torch.distributed.run with 2 ranks on a 1-GPU box, BOTH
ranks bound to cuda:0. NCCL/RCCL communicators normally reject two ranks
sharing one device; this probe records whichever happens — a working
init + fused forest all-gather + allreduce over the real RCCL backend,
or the library's refusal — as hardware evidence for profiles/. A 2-GPU
lease would run the same code with LOCAL_RANK binding unchanged.
"""

import datetime
import os
import sys

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    n_dev = torch.cuda.device_count()
    dev = min(int(os.environ.get("LOCAL_RANK", rank)), n_dev - 1)
    torch.cuda.set_device(dev)
    print(f"[rank {rank}] devices={n_dev} using cuda:{dev}", flush=True)
    try:
        dist.init_process_group(
            "nccl", timeout=datetime.timedelta(seconds=60))
        from isolation_forest_amd.parallel.comm import Comm

        comm = Comm()
        total = comm.all_reduce_sum_int(rank + 1)
        assert total == world * (world + 1) // 2, total
        # the fused SoA forest all-gather (the fit-path collective)
        T = 3 + rank
        arrs = comm.all_gather_forest_arrays(
            feature=np.full((T, 7), rank, dtype=np.int32),
            value=np.full((T, 7), float(rank), dtype=np.float32),
        )
        assert arrs["feature"].shape[0] == sum(3 + r for r in range(world))
        x = torch.ones(1024, device=f"cuda:{dev}")
        comm.all_reduce(x)
        assert float(x[0]) == world
        comm.barrier()
        if rank == 0:
            print(f"RCCL_OK world={world} backend={dist.get_backend()} "
                  f"allgather_trees={arrs['feature'].shape[0]}", flush=True)
    except Exception as e:  # noqa: BLE001 — the refusal IS the evidence
        print(f"RCCL_FAIL rank={rank}: {type(e).__name__}: {e}", flush=True)
        sys.exit(3)


if __name__ == "__main__":
    main()
