"""Dump the GPU-built extended forest for the failing parity case."""
import sys

import numpy as np
import torch

sys.path.insert(0, ".")
from isolation_forest_amd.core import cpu_engine
from isolation_forest_amd.ops import gpu_engine
from isolation_forest_amd.utils.params import ResolvedParams

n, d, T = 256, 8, 8
ext = int(sys.argv[1]) if len(sys.argv) > 1 else 7
X = np.random.RandomState(3).normal(size=(6000, d)).astype(np.float32)
bag = cpu_engine.sample_bags(6000, T, n, seed=4, bootstrap=False)
fs = cpu_engine.feature_subsets(d, d, T, seed=4)
rp = ResolvedParams(num_samples=n, num_features=d, total_rows=6000,
                    total_features=d, extension_level=ext)
g = gpu_engine.build_extended_forest(torch.from_numpy(X).to("cuda"), bag, fs, 4, rp)
np.savez(
    f"gpurun_out/eif_gpu_ext{ext}.npz",
    **{k: getattr(g, k) for k in [
        "feature", "value", "right", "num_instances", "node_count",
        "hyper_idx", "hyper_w", "offset64",
    ]},
)
print("saved ext", ext)
