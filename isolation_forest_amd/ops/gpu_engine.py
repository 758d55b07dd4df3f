"""GPU engine orchestration: device builds and scoring via the HIP extension.

The canonical forest lives in host numpy (core/forest.py) — it is small
(MBs) next to the data (GBs-TBs); device copies are cached per model and
device. Bags/feature subsets are drawn host-side with the same Philox
counters the kernels use, so a GPU-built forest is BIT-IDENTICAL to the
CPU oracle's for the same seed (tests/test_gpu.py enforces this).
"""

from __future__ import annotations

import math

import numpy as np
import torch

from ..core.forest import ExtendedForest, Forest
from ..utils.math import avg_path_length
from . import load_extension


def _leaf_lut(n: int, device) -> torch.Tensor:
    lut = avg_path_length(np.arange(n + 1))
    return torch.from_numpy(np.ascontiguousarray(lut, dtype=np.float32)).to(device)


def _mask_padding(feature, right, count, ncount, pad):
    cols = np.arange(feature.shape[1], dtype=np.int32)[None, :]
    live = cols < ncount[:, None]
    feature[~live] = pad
    right[~live] = -1
    count[~live] = -1
    return live


def build_forest(X: torch.Tensor, bag_idx: np.ndarray, feat_sub: np.ndarray,
                 seed: int, rp, tree_id_offset: int = 0) -> Forest:
    ext = load_extension()
    device = X.device
    n = rp.num_samples
    height_limit = int(math.ceil(math.log2(max(n, 2))))
    max_nodes = 2 * n - 1
    bag_t = torch.from_numpy(np.ascontiguousarray(bag_idx)).to(device)
    fs_t = torch.from_numpy(np.ascontiguousarray(feat_sub, dtype=np.int32)).to(device)
    bags = ext.bag_gather(X.contiguous(), bag_t)
    feat, value, right, count, ncount = ext.build_forest(
        bags, fs_t, seed, tree_id_offset, _leaf_lut(n, device), max_nodes,
        height_limit,
    )
    f_np = feat.cpu().numpy()
    v_np = value.cpu().numpy()
    r_np = right.cpu().numpy()
    c_np = count.cpu().numpy().astype(np.int64)
    nc_np = ncount.cpu().numpy()
    _mask_padding(f_np, r_np, c_np, nc_np, Forest.PAD)
    v_np = np.where(f_np >= 0, v_np, np.where(f_np == Forest.LEAF, v_np, 0.0)).astype(
        np.float32
    )
    return Forest(
        feature=f_np,
        value=v_np,
        right=r_np,
        num_instances=c_np,
        node_count=nc_np.astype(np.int32),
        num_samples=n,
        num_features=rp.num_features,
        total_num_features=rp.total_features,
        value64=np.where(f_np >= 0, v_np, 0.0).astype(np.float64),
    )


def build_extended_forest(X: torch.Tensor, bag_idx: np.ndarray,
                          feat_sub: np.ndarray, seed: int, rp,
                          tree_id_offset: int = 0) -> ExtendedForest:
    ext = load_extension()
    device = X.device
    n = rp.num_samples
    height_limit = int(math.ceil(math.log2(max(n, 2))))
    max_nodes = 2 ** (height_limit + 1) - 1
    nnz = min(rp.extension_level + 1, rp.num_features)
    bag_t = torch.from_numpy(np.ascontiguousarray(bag_idx)).to(device)
    fs_t = torch.from_numpy(np.ascontiguousarray(feat_sub, dtype=np.int32)).to(device)
    bags = ext.bag_gather(X.contiguous(), bag_t)
    feat, value, right, count, ncount, hidx, hw, off64 = ext.build_extended_forest(
        bags, fs_t, seed, tree_id_offset, _leaf_lut(n, device), nnz, max_nodes,
        height_limit,
    )
    f_np = feat.cpu().numpy()
    v_np = value.cpu().numpy()
    r_np = right.cpu().numpy()
    c_np = count.cpu().numpy().astype(np.int64)
    nc_np = ncount.cpu().numpy()
    live = _mask_padding(f_np, r_np, c_np, nc_np, ExtendedForest.PAD)
    v_np = np.where(live, v_np, 0.0).astype(np.float32)
    hidx_np = hidx.cpu().numpy()
    hw_np = hw.cpu().numpy()
    off64_np = off64.cpu().numpy()
    internal = (f_np >= 0) & live
    hidx_np[~internal] = 0
    hw_np[~internal] = 0.0
    off64_np[~internal] = 0.0
    return ExtendedForest(
        feature=f_np,
        value=v_np,
        right=r_np,
        num_instances=c_np,
        node_count=nc_np.astype(np.int32),
        hyper_idx=hidx_np,
        hyper_w=hw_np,
        offset64=off64_np,
        num_samples=n,
        num_features=rp.num_features,
        total_num_features=rp.total_features,
        extension_level=rp.extension_level,
    )


# ---------------------------------------------------------------------------
# scoring
# ---------------------------------------------------------------------------


def _nodes_packed(forest) -> np.ndarray:
    """Packed 8-byte node records for the scoring kernels:
    meta<0 => leaf (value = c(count)); else feature = meta&0xFFF,
    right = (meta>>12)&0x7FFF (left child is implicit pre-order id+1)."""
    T, mn = forest.feature.shape
    if mn > 32767:
        raise ValueError("forest too deep for the packed node format")
    feat = forest.feature
    if feat.max(initial=0) > 4095:
        raise ValueError("GPU scoring supports at most 4096 features")
    internal = feat >= 0
    meta = np.where(
        internal,
        feat.astype(np.int32) | (forest.right.astype(np.int32) << 12),
        np.int32(-2147483648),
    ).astype(np.int32)
    packed = np.empty((T, mn, 2), dtype=np.int32)
    packed[..., 0] = meta
    packed[..., 1] = forest.value.view(np.int32)
    return packed


def _device_forest(model, device):
    key = (str(device), id(model.forest))
    cache = model._gpu_forest_cache
    if key not in cache:
        forest = model.forest
        aos = torch.from_numpy(_nodes_packed(forest)).to(device)
        ncount = torch.from_numpy(
            np.ascontiguousarray(forest.node_count, dtype=np.int32)
        ).to(device)
        extra = {}
        if isinstance(forest, ExtendedForest):
            extra["hidx"] = torch.from_numpy(
                np.ascontiguousarray(forest.hyper_idx)
            ).to(device)
            extra["hw"] = torch.from_numpy(
                np.ascontiguousarray(forest.hyper_w)
            ).to(device)
        cache.clear()  # one cached device copy per model is enough
        cache[key] = (aos, ncount, extra)
    return cache[key]


def score_forest(model, X: torch.Tensor, finalize: bool = True) -> torch.Tensor:
    ext = load_extension()
    forest = model.forest
    aos, ncount, _ = _device_forest(model, X.device)
    c = float(avg_path_length(forest.num_samples))
    return ext.score_forest(X.contiguous(), aos, ncount, c, finalize)


def score_extended_forest(model, X: torch.Tensor, finalize: bool = True) -> torch.Tensor:
    ext = load_extension()
    forest = model.forest
    aos, ncount, extra = _device_forest(model, X.device)
    c = float(avg_path_length(forest.num_samples))
    return ext.score_extended_forest(
        X.contiguous(), aos, extra["hidx"], extra["hw"], ncount, c, finalize
    )
