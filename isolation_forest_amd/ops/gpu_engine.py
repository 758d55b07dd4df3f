"""GPU engine orchestration: device builds and scoring via the HIP extension.

The canonical forest lives in host numpy (core/forest.py) — it is small
(MBs) next to the data (GBs-TBs). Bags/feature subsets are drawn host-side
with the same Philox counters the kernels use, so a GPU-built forest is
BIT-IDENTICAL to the CPU oracle's for the same seed (tests/test_gpu.py
enforces this).

Scoring packs (the v4 integer-key node format, EIF dense/sparse variants)
are built ON DEVICE with torch ops from the raw build outputs (depths come
from the build kernels) and cached per model+device; the numpy host pack
is the fallback for loaded models and is bitwise-identical
(tests/test_gpu.py::TestDevicePacking). EIF scoring routes by hyperplane
width and dtype (score_extended_forest):

* nnz == 1 (extensionLevel 0) -> the STANDARD v4 walk via exact key
  thresholds + mirrored subtrees (_eif0_packed_v4; bitwise; NaN rows
  fall through to the strict kernels),
* nnz <= 5 uniform -> sparse v2 (bitwise strict j-order),
* densifiable (nnz == d or nnz >= 6): bf16 rows -> dense v3 (packed
  bf16 weights, d <= 128), f32 rows -> dense v2 (d <= 64) — tolerance
  contracts per PARITY.md,
* otherwise (wide d, ragged widths, LDS overflow) -> the general
  strict-order kernel; forests past the packed 15-bit-node/12-bit-
  feature caps use the wide int4 kernels.
"""

from __future__ import annotations

import math
import os

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
    feat, value, right, count, ncount, depth = ext.build_forest(
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
    forest = Forest(
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
    # raw device tensors for scoring-pack reuse (skips the host round trip);
    # depth_np additionally travels through the forest all-gather so
    # multi-rank fits keep the device packing path
    forest._device_raw = {
        "device": str(device), "feat": feat, "value": value, "right": right,
        "ncount": ncount, "depth": depth,
    }
    forest.depth_np = depth.cpu().numpy()
    return forest


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
    (feat, value, right, count, ncount, hidx, hw, off64,
     depth) = ext.build_extended_forest(
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
    fr = ExtendedForest(
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
    fr._device_raw = {
        "device": str(device), "feat": feat, "value": value, "right": right,
        "ncount": ncount, "depth": depth, "hidx": hidx, "hw": hw,
    }
    fr.depth_np = depth.cpu().numpy()
    return fr


# ---------------------------------------------------------------------------
# scoring
# ---------------------------------------------------------------------------


_BF16_KEY_TABLES = None


def _bf16_key_tables():
    """(values, keys) of every producible bf16 KEY, value-ascending.

    The key transform is the device's key16(): order-preserving u16 with
    +-0 collapsed to 0x8000 and NaN at 0xFFFF. 'Producible' = what the
    staging transform can emit for non-NaN input: finite and +-inf keys,
    excluding 0x7FFF (-0 maps to +0's key)."""
    global _BF16_KEY_TABLES
    if _BF16_KEY_TABLES is None:
        keys = np.arange(65536, dtype=np.uint32)
        pos = keys >= 0x8000
        bits = np.where(pos, keys - 0x8000,
                        0x8000 + (0x7FFF - keys)).astype(np.uint32)
        vals = (bits << np.uint32(16)).view(np.float32)
        ok = np.isfinite(vals) | np.isinf(vals)
        ok[0x7FFF] = False  # -0's raw key: never staged
        _BF16_KEY_TABLES = (vals[ok].copy(), keys[ok].astype(np.uint32))
    return _BF16_KEY_TABLES


def _key32(vals: np.ndarray) -> np.ndarray:
    """Order-preserving u32 key of f32 values (device key32())."""
    b = np.ascontiguousarray(vals, dtype=np.float32).view(np.uint32)
    m = b & np.uint32(0x7FFFFFFF)
    k = np.where(b & np.uint32(0x80000000),
                 np.uint32(0x7FFFFFFF) - m,
                 np.uint32(0x80000000) + m).astype(np.uint32)
    k[m == 0] = np.uint32(0x80000000)
    k[m > np.uint32(0x7F800000)] = np.uint32(0xFFFFFFFF)
    return k


def _bf16_threshold_keys_table(split_vals: np.ndarray) -> np.ndarray:
    """Table/searchsorted reference implementation (used by tests to
    cross-check the closed form below)."""
    vals, keys = _bf16_key_tables()
    idx = np.searchsorted(vals, split_vals.astype(np.float32), side="left")
    return keys[idx].astype(np.uint32) << np.uint32(16)


def _bf16_threshold_keys(split_vals: np.ndarray) -> np.ndarray:
    """Smallest producible bf16 key whose value is >= the f32 split value,
    shifted into the high 16 bits (kernel compares widened keys).

    Closed form: for s > 0 the smallest bf16 >= s is the truncated top-16
    bits, +1 when any mantissa bits were cut (carry into the exponent /
    into +inf is exactly right); for s <= 0 truncation already rounds
    toward +inf. Then apply the device key16() transform."""
    bits = np.ascontiguousarray(split_vals, dtype=np.float32).view(np.uint32)
    neg = (bits & np.uint32(0x80000000)) != 0
    b = (bits >> np.uint32(16)).astype(np.uint32)
    frac = (bits & np.uint32(0xFFFF)) != 0
    b = np.where(~neg & frac, b + np.uint32(1), b).astype(np.uint32)
    # device key16(): +-0 -> 0x8000, monotone elsewhere (s is never NaN)
    m = b & np.uint32(0x7FFF)
    k = np.where(b & np.uint32(0x8000),
                 np.uint32(0x7FFF) - m,
                 np.uint32(0x8000) + m).astype(np.uint32)
    k[m == 0] = np.uint32(0x8000)
    return k << np.uint32(16)


def _node_depths(feature: np.ndarray, right: np.ndarray) -> np.ndarray:
    """Per-node depth via a level-synchronous frontier sweep (one vectorized
    gather/scatter per level, ~height iterations total)."""
    T, mn = feature.shape
    depth = np.zeros((T, mn), dtype=np.int32)
    ft = np.repeat(np.arange(T, dtype=np.int64), 1)
    fn = np.zeros(T, dtype=np.int64)  # frontier: root of every tree
    level = 0
    while len(ft):
        internal = feature[ft, fn] >= 0
        ft, fn = ft[internal], fn[internal]
        if not len(ft):
            break
        level += 1
        rt = right[ft, fn].astype(np.int64)
        ft = np.concatenate([ft, ft])
        fn = np.concatenate([fn + 1, rt])
        depth[ft, fn] = level
    return depth


def _split_thresholds32(forest) -> np.ndarray:
    """f32 thresholds implementing the EXACT f64 split compare.

    The reference compares (double)(float x) < splitValue with splitValue
    a double (IsolationTree.scala:213-229). For f32 x and f64 s:
    x < s  <=>  x < ceil32(s) (the smallest f32 >= s), because no f32 lies
    strictly between two f32 neighbours. Our own builds store f32-exact
    splits (value64 == f32 value), so ceil32 is the identity there; loaded
    foreign models get the exact Spark semantics."""
    v64 = (forest.value64 if getattr(forest, "value64", None) is not None
           else forest.value.astype(np.float64))
    v32 = v64.astype(np.float32)
    lift = v32.astype(np.float64) < v64
    return np.where(lift, np.nextafter(v32, np.float32(np.inf)),
                    v32).astype(np.float32)


def _nodes_packed_v4(forest, d_sentinel: int, bf16: bool):
    """v4 packed nodes for score_forest_v4 (see forest_kernels.hip):
    w0 = feat | right<<12; internal w1 = integer key threshold; leaf:
    feat = d_sentinel, right = own id (self-loop), w1 = f32(depth + c(m)).
    Trees are padded to a multiple of 8 with single-leaf value-0 dummies
    (bindings.cpp enforces Tpad % 8 == 0).
    Returns (packed int32 [Tpad, mn, 2], ncount int32 [Tpad])."""
    T, mn = forest.feature.shape
    if mn > 32767:
        raise ValueError("forest too deep for the packed node format")
    feat = forest.feature
    if feat.max(initial=0) > 4094 or d_sentinel > 4094:
        raise ValueError("GPU scoring supports at most 4095 features")
    right = forest.right
    val = forest.value
    internal = feat >= 0
    leaf = feat == Forest.LEAF
    depth = _node_depths(feat, right)
    ids = np.broadcast_to(np.arange(mn, dtype=np.int32)[None, :], (T, mn))
    w0 = np.where(
        internal,
        feat.astype(np.int32) | (right.astype(np.int32) << 12),
        np.int32(d_sentinel) | (ids << 12),
    ).astype(np.int32)
    w1 = np.zeros((T, mn), dtype=np.uint32)
    if internal.any():
        s = _split_thresholds32(forest)[internal]
        w1[internal] = _bf16_threshold_keys(s) if bf16 else _key32(s)
    leafval = (depth.astype(np.float32) + val.astype(np.float32))
    w1[leaf] = leafval[leaf].astype(np.float32).view(np.uint32)
    pad = (-T) % 8
    if pad:
        w0p = np.empty((pad, mn), dtype=np.int32)
        w0p[:] = np.int32(d_sentinel) | (ids[0] << 12)
        w0 = np.concatenate([w0, w0p])
        w1 = np.concatenate([w1, np.zeros((pad, mn), dtype=np.uint32)])
    packed = np.empty((T + pad, mn, 2), dtype=np.int32)
    packed[..., 0] = w0
    packed[..., 1] = w1.view(np.int32)
    ncount = np.concatenate(
        [forest.node_count.astype(np.int32), np.ones(pad, dtype=np.int32)]
    )
    live = internal | leaf
    max_depth = int(depth[live].max()) if live.any() else 0
    return packed, ncount, max(max_depth, 1)


def _nodes_packed_wide(forest, bf16: bool):
    """Wide int4 node records {feat, right, key_or_value, 0} for
    score_forest_wide — no 12-bit feature / 15-bit node-id limits
    (the reference has no such size caps,
    IsolationTree.scala). Same key/leaf-value math as _nodes_packed_v4,
    so the walk stays bitwise vs cpu_engine.path_lengths. Leaves and pad
    nodes self-loop (right = own id)."""
    T, mn = forest.feature.shape
    feat = forest.feature
    right = forest.right
    val = forest.value
    internal = feat >= 0
    leaf = feat == Forest.LEAF
    depth = _node_depths(feat, right)
    ids = np.broadcast_to(np.arange(mn, dtype=np.int32)[None, :], (T, mn))
    w0 = np.where(internal, feat.astype(np.int32), np.int32(-1))
    w1 = np.where(internal, right.astype(np.int32), ids).astype(np.int32)
    w2 = np.zeros((T, mn), dtype=np.uint32)
    if internal.any():
        s = _split_thresholds32(forest)[internal]
        keys = _bf16_threshold_keys(s) if bf16 else _key32(s)
        w2[internal] = keys.astype(np.uint32)
    leafval = depth.astype(np.float32) + val.astype(np.float32)
    w2[leaf] = leafval[leaf].astype(np.float32).view(np.uint32)
    packed = np.zeros((T, mn, 4), dtype=np.int32)
    packed[..., 0] = w0
    packed[..., 1] = w1
    packed[..., 2] = w2.view(np.int32)
    live = internal | leaf
    max_depth = int(depth[live].max()) if live.any() else 0
    return packed, max(max_depth, 1)


def _eif_packed_wide(forest):
    """Wide int4 node records {is_leaf(-1)/0, right, value_bits, 0} for
    score_extended_wide (strict oracle j-order dot from global
    hyperplanes)."""
    T, mn = forest.feature.shape
    feat = forest.feature
    internal = feat >= 0
    ids = np.broadcast_to(np.arange(mn, dtype=np.int32)[None, :], (T, mn))
    packed = np.zeros((T, mn, 4), dtype=np.int32)
    packed[..., 0] = np.where(internal, np.int32(0), np.int32(-1))
    packed[..., 1] = np.where(internal, forest.right.astype(np.int32), ids)
    packed[..., 2] = forest.value.astype(np.float32).view(np.int32)
    return packed


def _key32_inverse(k: np.ndarray) -> np.ndarray:
    """Inverse of _key32 over the finite/inf key range (value of a key)."""
    k = np.asarray(k, dtype=np.uint32)
    pos = k >= np.uint32(0x80000000)
    bits = np.where(pos, k - np.uint32(0x80000000),
                    np.uint32(0x80000000) | (np.uint32(0x7FFFFFFF) - k))
    return bits.astype(np.uint32).view(np.float32)


def _eif0_eligible(forest, d: int) -> bool:
    """extensionLevel-0 fast-path eligibility: every internal node has one
    finite non-zero weight and a 12-bit coordinate (our builds give w=+-1;
    foreign single-coordinate files qualify too)."""
    flag = getattr(forest, "_eif0_ok", None)
    if flag is None:
        internal = forest.feature >= 0
        w = forest.hyper_w[..., 0]
        coord = forest.hyper_idx[..., 0]
        flag = bool(
            d <= 4094
            and (not internal.any()
                 or (np.isfinite(w[internal]).all()
                     and (w[internal] != 0).all()
                     and coord[internal].max(initial=0) <= 4094))
        )
        forest._eif0_ok = flag
    return flag


def _eif0_packed_v4(forest, d_sentinel: int, bf16: bool):
    """Pack an extensionLevel-0 EIF forest into STANDARD v4 node records,
    so scoring reuses the unmodified fixed-trip walk (score_forest_v4)
    with NO per-visit multiply and NO extra compare.

    Two exact transformations compose:

    1. Thresholds. The oracle predicate per internal node is
       P(x) = f32(w*x) < offset32 (cpu_engine.path_lengths_extended,
       nnz=1). f32 multiplication by a fixed finite non-zero w is
       monotone in x, so P is a threshold predicate along the
       order-preserving key space: a vectorized binary search (evaluating
       P with the same numpy f32 multiply the oracle uses) finds, per
       node, the exact cut key K with
           w > 0:  P(x) <=> key(x) <  K
           w < 0:  P(x) <=> key(x) >= K
    2. Subtree mirroring. Negative-weight nodes (w < 0) have their left
       and right SUBTREES swapped in a re-materialized pre-order layout,
       turning "go left iff key >= K" into the v4 kernel's native
       "go left iff key < K" (the compare-false branch is the old left).
       Node depths — the only structural quantity scoring uses — are
       invariant under mirroring.

    The GPU performs no arithmetic on w at all, so the route is bitwise
    vs the oracle BY CONSTRUCTION for every representable non-NaN x.
    NaN features are the one exception (the oracle's NaN dot compares
    false at EVERY node, which no single mirrored threshold can encode):
    callers route rows containing NaN to the strict-order kernels
    (score_extended_forest checks torch.isnan once per call)."""
    T, mn = forest.feature.shape
    feat = forest.feature
    internal = feat >= 0
    leaf = feat == ExtendedForest.LEAF
    depth = _node_depths(feat, forest.right)
    w = forest.hyper_w[..., 0].astype(np.float32)
    coord = forest.hyper_idx[..., 0].astype(np.int32)
    o = forest.value.astype(np.float32)  # internal: offset32
    flip = (w < 0) & internal

    iw = w[internal]
    io = o[internal]
    iflip = flip[internal]

    if bf16:
        vals, keys = _bf16_key_tables()
        lo = np.zeros(iw.shape, dtype=np.int64)
        hi = np.full(iw.shape, len(vals), dtype=np.int64)
        for _ in range(18):
            mid = (lo + hi) // 2
            midc = np.minimum(mid, len(vals) - 1)
            pm = (iw * vals[midc]).astype(np.float32) < io
            qm = np.where(iflip, ~pm, pm) & (mid < len(vals))
            lo = np.where(qm, mid + 1, lo)
            hi = np.where(qm, hi, mid)
        k16 = np.where(hi < len(vals), keys[np.minimum(hi, len(vals) - 1)],
                       np.int64(keys[-1]) + 1).astype(np.int64)
        K = (k16 << 16).astype(np.uint64).astype(np.uint32)
    else:
        kmin = int(_key32(np.array([-np.inf], dtype=np.float32))[0])
        kmax = int(_key32(np.array([np.inf], dtype=np.float32))[0])
        lo = np.full(iw.shape, kmin, dtype=np.int64)
        hi = np.full(iw.shape, kmax + 1, dtype=np.int64)
        for _ in range(33):
            mid = (lo + hi) // 2
            xm = _key32_inverse(np.minimum(mid, kmax).astype(np.uint64))
            pm = (iw * xm).astype(np.float32) < io
            qm = np.where(iflip, ~pm, pm) & (mid <= kmax)
            lo = np.where(qm, mid + 1, lo)
            hi = np.where(qm, hi, mid)
        K = hi.astype(np.uint64).astype(np.uint32)

    # --- mirrored pre-order re-layout (level-synchronous, vectorized) ---
    right = forest.right.astype(np.int64)
    nc = forest.node_count.astype(np.int64)
    newpos = np.zeros((T, mn), dtype=np.int64)
    end_old = np.zeros((T, mn), dtype=np.int64)  # old-coords subtree end
    end_old[np.arange(T), 0] = nc
    new_rightpos = np.zeros((T, mn), dtype=np.int64)
    ft = np.arange(T, dtype=np.int64)
    fn = np.zeros(T, dtype=np.int64)
    while len(ft):
        isint = internal[ft, fn]
        ft, fn = ft[isint], fn[isint]
        if not len(ft):
            break
        rt = right[ft, fn]
        s_left = rt - (fn + 1)
        s_right = end_old[ft, fn] - rt
        fl = flip[ft, fn]
        lpos = newpos[ft, fn] + 1
        # new left child = old right when flipped (compare-true branch)
        newpos[ft, fn + 1] = np.where(fl, lpos + s_right, lpos)
        newpos[ft, rt] = np.where(fl, lpos, lpos + s_left)
        new_rightpos[ft, fn] = np.where(fl, newpos[ft, fn + 1],
                                        newpos[ft, rt])
        end_old[ft, fn + 1] = rt
        end_old[ft, rt] = end_old[ft, fn]
        ft = np.concatenate([ft, ft])
        fn = np.concatenate([fn + 1, rt])

    ids = np.broadcast_to(np.arange(mn, dtype=np.int32)[None, :], (T, mn))
    w0 = np.empty((T, mn), dtype=np.int32)
    w0[:] = np.int32(d_sentinel) | (ids << 12)  # pad slots: self-loop
    w1 = np.zeros((T, mn), dtype=np.uint32)
    ti = np.broadcast_to(np.arange(T, dtype=np.int64)[:, None], (T, mn))
    # scatter live nodes to their mirrored positions
    it_, in_ = ti[internal], newpos[internal]
    w0[it_, in_] = (coord[internal]
                    | (new_rightpos[internal].astype(np.int32) << 12))
    w1[it_, in_] = K
    lt_, ln_ = ti[leaf], newpos[leaf]
    w0[lt_, ln_] = np.int32(d_sentinel) | (ln_.astype(np.int32) << 12)
    leafval = (depth.astype(np.float32)
               + forest.value.astype(np.float32))[leaf]
    w1[lt_, ln_] = leafval.astype(np.float32).view(np.uint32)

    pad = (-T) % 8
    if pad:
        w0p = np.empty((pad, mn), dtype=np.int32)
        w0p[:] = np.int32(d_sentinel) | (ids[0] << 12)
        w0 = np.concatenate([w0, w0p])
        w1 = np.concatenate([w1, np.zeros((pad, mn), dtype=np.uint32)])
    packed = np.empty((T + pad, mn, 2), dtype=np.int32)
    packed[..., 0] = w0
    packed[..., 1] = w1.view(np.int32)
    ncount = np.concatenate(
        [forest.node_count.astype(np.int32), np.ones(pad, dtype=np.int32)]
    )
    live = internal | leaf
    max_depth = int(depth[live].max()) if live.any() else 0
    return packed, ncount, max(max_depth, 1)


def _densify_weights(hidx: np.ndarray, hw: np.ndarray, counts: np.ndarray,
                     D: int) -> np.ndarray:
    """Scatter sparse hyperplanes [T, mn, nnz] into dense rows [T, mn, D].
    Padded slots (j >= counts) are redirected to a scratch column so their
    zero index cannot clobber a real coordinate-0 weight."""
    T, mn, nnz = hidx.shape
    j = np.arange(nnz, dtype=np.int64)[None, None, :]
    idx = np.where(j < counts[:, :, None], hidx.astype(np.int64), D)
    dense = np.zeros((T, mn, D + 1), dtype=np.float32)
    np.put_along_axis(dense, idx, hw.astype(np.float32), axis=2)
    return np.ascontiguousarray(dense[:, :, :D])


def _eif_nodes_values(forest):
    """Shared EIF v2 node packing: w0 = right<<12 (leaf/pad: own id ->
    self-loop), w1 = offset f32 (leaf/pad: -inf), depth-folded leaf values.
    Returns (packed int32 [T, mn, 2], values f32 [T, mn], max_depth)."""
    T, mn = forest.feature.shape
    if mn > 32767:
        raise ValueError("forest too deep for the packed node format")
    feat = forest.feature
    internal = feat >= 0
    leaf = feat == ExtendedForest.LEAF
    depth = _node_depths(feat, forest.right)
    ids = np.broadcast_to(np.arange(mn, dtype=np.int32)[None, :], (T, mn))
    w0 = np.where(internal, forest.right.astype(np.int32), ids) << 12
    w1 = np.where(
        internal,
        forest.value.astype(np.float32),
        np.float32(-np.inf),
    ).astype(np.float32)
    packed = np.empty((T, mn, 2), dtype=np.int32)
    packed[..., 0] = w0
    packed[..., 1] = w1.view(np.int32)
    values = np.where(
        leaf,
        depth.astype(np.float32) + forest.value.astype(np.float32),
        np.float32(0.0),
    ).astype(np.float32)
    live = internal | leaf
    max_depth = int(depth[live].max()) if live.any() else 0
    return packed, values, max(max_depth, 1)


def _eif_dense_packed(forest, D: int):
    """EIF v2 packing + DENSIFIED weight matrix (sparse hyperplanes
    scattered to D columns) for score_extended_dense_v2."""
    packed, values, max_depth = _eif_nodes_values(forest)
    feat = forest.feature
    counts = np.where(feat >= 0, feat, 0).astype(np.int64)
    hw_dense = _densify_weights(forest.hyper_idx, forest.hyper_w, counts, D)
    return packed, values, hw_dense, max_depth


def _nodes_packed(forest) -> np.ndarray:
    """Packed 8-byte node records for the EIF scoring kernels:
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


_TWO32 = 1 << 32
_TWO31 = 1 << 31


def _to_i32(x64: torch.Tensor) -> torch.Tensor:
    """int64 (holding u32 bit patterns) -> int32 with C-style wraparound."""
    return torch.where(x64 >= _TWO31, x64 - _TWO32, x64).to(torch.int32)


def _bf16_threshold_keys_dev(vals_f32: torch.Tensor) -> torch.Tensor:
    """Device mirror of _bf16_threshold_keys: int64 holding (t16 << 16)."""
    bits = vals_f32.view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    neg = (bits & 0x80000000) != 0
    b = bits >> 16
    frac = (bits & 0xFFFF) != 0
    b = torch.where(~neg & frac, b + 1, b)
    m = b & 0x7FFF
    k = torch.where((b & 0x8000) != 0, 0x7FFF - m, 0x8000 + m)
    k = torch.where(m == 0, torch.full_like(k, 0x8000), k)
    return k << 16


def _key32_dev(vals_f32: torch.Tensor) -> torch.Tensor:
    """Device mirror of _key32: int64 holding u32 keys (inputs never NaN)."""
    b = vals_f32.view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    m = b & 0x7FFFFFFF
    k = torch.where((b & 0x80000000) != 0, 0x7FFFFFFF - m, 0x80000000 + m)
    return torch.where(m == 0, torch.full_like(k, 0x80000000), k)


def _nodes_packed_v4_device(raw, num_trees: int, d_sentinel: int, bf16: bool):
    """Torch-side v4 packing from the raw build outputs still on device —
    identical bit layout to _nodes_packed_v4, no host round trip."""
    feat, value = raw["feat"], raw["value"]
    right, ncount, depth = raw["right"], raw["ncount"], raw["depth"]
    T, mn = feat.shape
    dev = feat.device
    ids = torch.arange(mn, dtype=torch.int32, device=dev).expand(T, mn)
    live = ids < ncount.to(torch.int32).unsqueeze(1)
    internal = live & (feat >= 0)
    leaf = live & (feat == -1)
    w0 = torch.where(
        internal,
        feat | (right << 12),
        torch.tensor(d_sentinel, dtype=torch.int32, device=dev) | (ids << 12),
    )
    tkeys = (_bf16_threshold_keys_dev(value) if bf16 else _key32_dev(value))
    leafbits = (depth.to(torch.float32) + value).view(torch.int32)
    w1 = torch.where(internal, _to_i32(tkeys),
                     torch.where(leaf, leafbits,
                                 torch.zeros_like(leafbits)))
    pad = (-T) % 8
    if pad:
        w0p = (torch.tensor(d_sentinel, dtype=torch.int32, device=dev)
               | (ids[:1] << 12)).expand(pad, mn).contiguous()
        w0 = torch.cat([w0, w0p])
        w1 = torch.cat([w1, torch.zeros(pad, mn, dtype=torch.int32,
                                        device=dev)])
    packed = torch.stack([w0, w1], dim=2).contiguous()
    ncount_p = torch.cat([
        ncount.to(torch.int32),
        torch.ones(pad, dtype=torch.int32, device=dev)])
    max_depth = int(depth.masked_fill(~live, 0).max().item())
    return packed, ncount_p, max(max_depth, 1)


def _eif_nodes_values_device(raw):
    """Torch-side mirror of _eif_nodes_values."""
    feat, value = raw["feat"], raw["value"]
    right, ncount, depth = raw["right"], raw["ncount"], raw["depth"]
    T, mn = feat.shape
    dev = feat.device
    ids = torch.arange(mn, dtype=torch.int32, device=dev).expand(T, mn)
    live = ids < ncount.to(torch.int32).unsqueeze(1)
    internal = live & (feat >= 0)
    leaf = live & (feat == -1)
    w0 = torch.where(internal, right, ids) << 12
    ninf = torch.tensor(float("-inf"), device=dev)
    w1 = torch.where(internal, value, ninf).view(torch.int32)
    packed = torch.stack([w0, w1], dim=2).contiguous()
    values = torch.where(leaf, depth.to(torch.float32) + value,
                         torch.zeros_like(value)).contiguous()
    max_depth = int(depth.masked_fill(~live, 0).max().item())
    return packed, values, internal, max(max_depth, 1)


def _eif_dense_packed_device(raw, D: int):
    """Torch-side mirror of _eif_dense_packed (densified weights)."""
    packed, values, internal, max_depth = _eif_nodes_values_device(raw)
    feat, hidx, hw = raw["feat"], raw["hidx"], raw["hw"]
    T, mn = feat.shape
    dev = feat.device
    counts = torch.where(internal, feat, torch.zeros_like(feat)).to(torch.int64)
    nnz = hidx.shape[2]
    j = torch.arange(nnz, dtype=torch.int64, device=dev).view(1, 1, nnz)
    idx = torch.where(j < counts.unsqueeze(2), hidx.to(torch.int64),
                      torch.tensor(D, dtype=torch.int64, device=dev))
    dense = torch.zeros(T, mn, D + 1, dtype=torch.float32, device=dev)
    dense.scatter_(2, idx, hw)
    hw_dense = dense[:, :, :D].contiguous()
    return packed, values, hw_dense, max_depth


def _device_forest(model, device, v4_key=None):
    """Cached device copy of the packed forest. v4_key = (d, bf16) selects
    the standard-scoring v4 packing; None = the EIF packing."""
    key = (str(device), id(model.forest), v4_key)
    cache = model._gpu_forest_cache
    if key not in cache:
        forest = model.forest
        extra = {}
        raw = getattr(forest, "_device_raw", None)
        if raw is not None and raw["device"] != str(device):
            raw = None
        if raw is None and device.type == "cuda" and \
                getattr(forest, "depth_np", None) is not None:
            # e.g. an all-gathered multi-rank forest: rebuild the raw view
            # from host arrays (one upload) so packing stays on device
            raw = {
                "device": str(device),
                "feat": torch.from_numpy(
                    np.ascontiguousarray(forest.feature)).to(device),
                "value": torch.from_numpy(
                    np.ascontiguousarray(forest.value)).to(device),
                "right": torch.from_numpy(
                    np.ascontiguousarray(forest.right)).to(device),
                "ncount": torch.from_numpy(np.ascontiguousarray(
                    forest.node_count, dtype=np.int32)).to(device),
                "depth": torch.from_numpy(np.ascontiguousarray(
                    forest.depth_np, dtype=np.int32)).to(device),
            }
            if isinstance(forest, ExtendedForest):
                raw["hidx"] = torch.from_numpy(
                    np.ascontiguousarray(forest.hyper_idx)).to(device)
                raw["hw"] = torch.from_numpy(
                    np.ascontiguousarray(forest.hyper_w)).to(device)
            forest._device_raw = raw
        if v4_key == "eif_sparse":
            if raw is not None:
                aos, values_t, _, max_depth = _eif_nodes_values_device(raw)
                ncount = raw["ncount"]
                extra["values"] = values_t
                extra["hidx"] = raw["hidx"]
                extra["hw"] = raw["hw"]
            else:
                packed, values, max_depth = _eif_nodes_values(forest)
                aos = torch.from_numpy(packed).to(device)
                ncount = torch.from_numpy(
                    np.ascontiguousarray(forest.node_count, dtype=np.int32)
                ).to(device)
                extra["values"] = torch.from_numpy(values).to(device)
                extra["hidx"] = torch.from_numpy(
                    np.ascontiguousarray(forest.hyper_idx)).to(device)
                extra["hw"] = torch.from_numpy(
                    np.ascontiguousarray(forest.hyper_w)).to(device)
            extra["height"] = max_depth
        elif isinstance(v4_key, tuple) and v4_key[0] in ("eif_dense",
                                                         "eif_dense3"):
            D = v4_key[1]
            if raw is not None:
                aos, values_t, hw_t, max_depth = _eif_dense_packed_device(
                    raw, D)
                ncount = raw["ncount"]
                extra["values"] = values_t
                extra["hw"] = hw_t
            else:
                packed, values, hw_dense, max_depth = _eif_dense_packed(
                    forest, D)
                aos = torch.from_numpy(packed).to(device)
                ncount = torch.from_numpy(
                    np.ascontiguousarray(forest.node_count, dtype=np.int32)
                ).to(device)
                extra["values"] = torch.from_numpy(values).to(device)
                extra["hw"] = torch.from_numpy(hw_dense).to(device)
            if v4_key[0] == "eif_dense3":
                # v3 weight staging: RNE-round the trained f32 weights to
                # bf16 and pack pairs (j even low, j+1 high) into int32 for
                # the v_dot2c_f32_bf16 dot (PARITY.md EIF bf16 contract)
                hb = extra["hw"].to(torch.bfloat16).view(torch.uint16)
                hb = hb.to(torch.int32)
                extra["hwp"] = (
                    hb[..., 0::2] | (hb[..., 1::2] << 16)
                ).contiguous()
            extra["height"] = max_depth
        elif isinstance(v4_key, tuple) and v4_key[0] == "eif0":
            packed, ncount_np, max_depth = _eif0_packed_v4(
                forest, v4_key[1], v4_key[2])
            aos = torch.from_numpy(packed).to(device)
            ncount = torch.from_numpy(ncount_np).to(device)
            extra["height"] = max_depth
        elif isinstance(v4_key, tuple) and v4_key[0] == "wide":
            packed, max_depth = _nodes_packed_wide(forest, v4_key[1])
            aos = torch.from_numpy(packed).to(device)
            ncount = torch.from_numpy(
                np.ascontiguousarray(forest.node_count, dtype=np.int32)
            ).to(device)
            extra["height"] = max_depth
        elif v4_key == "eif_wide":
            packed = _eif_packed_wide(forest)
            aos = torch.from_numpy(packed).to(device)
            ncount = torch.from_numpy(
                np.ascontiguousarray(forest.node_count, dtype=np.int32)
            ).to(device)
            extra["hidx"] = torch.from_numpy(
                np.ascontiguousarray(forest.hyper_idx)).to(device)
            extra["hw"] = torch.from_numpy(
                np.ascontiguousarray(forest.hyper_w)).to(device)
        elif v4_key is not None:
            d, bf16 = v4_key
            if raw is not None:
                aos, ncount, max_depth = _nodes_packed_v4_device(
                    raw, forest.num_trees, d, bf16)
            else:
                packed, ncount_np, max_depth = _nodes_packed_v4(forest, d, bf16)
                aos = torch.from_numpy(packed).to(device)
                ncount = torch.from_numpy(ncount_np).to(device)
            extra["height"] = max_depth
        else:
            aos = torch.from_numpy(_nodes_packed(forest)).to(device)
            ncount = torch.from_numpy(
                np.ascontiguousarray(forest.node_count, dtype=np.int32)
            ).to(device)
            if isinstance(forest, ExtendedForest):
                extra["hidx"] = torch.from_numpy(
                    np.ascontiguousarray(forest.hyper_idx)
                ).to(device)
                extra["hw"] = torch.from_numpy(
                    np.ascontiguousarray(forest.hyper_w)
                ).to(device)
        # bounded FIFO (a few packings per model: e.g. alternating bf16/f32
        # scoring of one model keeps both — VERDICT r01 weak #6)
        while len(cache) >= 4:
            cache.pop(next(iter(cache)))
        cache[key] = (aos, ncount, extra)
    return cache[key]


def _packed_fits(forest, d: int) -> bool:
    """True when the packed v4/EIF formats can represent this forest
    (12-bit feature ids, 15-bit node ids)."""
    mn = forest.feature.shape[1]
    if mn > 32767:
        return False
    return int(forest.feature.max(initial=0)) <= 4094 and d <= 4094


def score_forest(model, X: torch.Tensor, finalize: bool = True) -> torch.Tensor:
    ext = load_extension()
    forest = model.forest
    d = int(X.shape[1])
    bf16 = X.dtype == torch.bfloat16
    c = float(avg_path_length(forest.num_samples))
    if not _packed_fits(forest, d):
        # loaded/CPU-built forests past the packed-format caps: wide int4
        # records, no field-width limits
        aos, ncount, extra = _device_forest(
            model, X.device, v4_key=("wide", bf16))
        return ext.score_forest_wide(
            X.contiguous(), aos, extra["height"], c, finalize,
        )
    aos, ncount, extra = _device_forest(model, X.device, v4_key=(d, bf16))
    return ext.score_forest(
        X.contiguous(), aos, ncount, forest.num_trees, extra["height"], c,
        finalize,
    )


def _eif_uniform_nnz(forest) -> bool:
    """True when every internal node carries exactly forest.nnz coordinates
    (always the case for forests we build; foreign ragged files route to the
    general kernel so trailing zero-terms cannot perturb the strict-order
    dot)."""
    u = getattr(forest, "_uniform_nnz", None)
    if u is None:
        f = forest.feature
        internal = f >= 0
        u = bool((f[internal] == forest.nnz).all()) if internal.any() else True
        forest._uniform_nnz = u
    return u


def score_extended_forest(model, X: torch.Tensor, finalize: bool = True) -> torch.Tensor:
    ext = load_extension()
    forest = model.forest
    c = float(avg_path_length(forest.num_samples))
    d = int(X.shape[1])
    nnz = forest.nnz
    mn = forest.feature.shape[1]
    if mn > 32767 or int(forest.feature.max(initial=0)) > 4094:
        # past the packed 15-bit node-id / 12-bit count caps: wide records
        aos, ncount, extra = _device_forest(model, X.device,
                                            v4_key="eif_wide")
        return ext.score_extended_wide(
            X.contiguous(), aos, extra["hidx"], extra["hw"], c, finalize,
        )
    # routing (measured crossovers, profiles/r01_bench_and_kernels.md):
    # nnz <= 5 with uniform hyperplane widths -> fixed-trip sparse v2;
    # wider hyperplanes with d <= 32 -> densified dense v2 walk;
    # everything else -> the general strict-order kernel.
    if (nnz == 1 and mn <= 32767 and _eif_uniform_nnz(forest)
            and _eif0_eligible(forest, d)
            and os.environ.get("IFA_EIF0_SPARSE") != "1"
            and not bool(torch.isnan(X).any().item())):
        # extensionLevel-0: exact key thresholds + mirrored subtrees ->
        # the UNMODIFIED standard v4 walk (no per-visit multiply; bitwise
        # by construction for non-NaN rows — NaN rows take the
        # strict-order route below, whose dot reproduces the oracle's
        # NaN-compares-false behaviour)
        aos, ncount, extra = _device_forest(
            model, X.device, v4_key=("eif0", d, X.dtype == torch.bfloat16))
        return ext.score_forest(
            X.contiguous(), aos, ncount, forest.num_trees, extra["height"],
            c, finalize,
        )
    if nnz <= 5 and _eif_uniform_nnz(forest):
        elem = 2 if X.dtype == torch.bfloat16 else 4
        dpad = d
        if elem == 2:
            while dpad % 4 != 2:
                dpad += 1
        else:
            while dpad % 2 != 1:
                dpad += 1
        mn = forest.feature.shape[1]
        lds = mn * (12 + nnz * 8) + 2 * 256 * dpad * elem
        if lds <= 150 * 1024:
            aos, ncount, extra = _device_forest(
                model, X.device, v4_key="eif_sparse")
            return ext.score_extended_sparse_v2(
                X.contiguous(), aos, extra["values"], extra["hidx"],
                extra["hw"], ncount, extra["height"], c, finalize,
            )
    if d <= 128 and (nnz == d or nnz >= 6):
        if d <= 8:
            D = 8
        elif d <= 16:
            D = 16
        elif d <= 32:
            D = 32
        elif d <= 64:
            D = 64
        else:
            D = 128
        mn = forest.feature.shape[1]
        use_v3 = (X.dtype == torch.bfloat16
                  and os.environ.get("IFA_EIF_DENSE_V2") != "1")
        if use_v3:
            lds = mn * 12 + 16 + mn * (D // 8 + 1) * 16
            if lds <= 160 * 1024:
                aos, ncount, extra = _device_forest(
                    model, X.device, v4_key=("eif_dense3", D))
                return ext.score_extended_dense_v3(
                    X.contiguous(), aos, extra["values"], extra["hwp"],
                    ncount, extra["height"], c, finalize,
                )
        if d <= 64:  # v2's f32 staging exceeds LDS past D=64
            lds = mn * 12 + 16 + mn * (D // 4 + 1) * 16
            if lds <= 160 * 1024:
                aos, ncount, extra = _device_forest(
                    model, X.device, v4_key=("eif_dense", D))
                return ext.score_extended_dense_v2(
                    X.contiguous(), aos, extra["values"], extra["hw"],
                    ncount, extra["height"], c, finalize,
                )
        # deep forests overflow the dense kernel's LDS weight staging:
        # fall through to the general kernel (nodes from global if needed)
    aos, ncount, extra = _device_forest(model, X.device)
    return ext.score_extended_forest(
        X.contiguous(), aos, extra["hidx"], extra["hw"], ncount, c, finalize
    )
