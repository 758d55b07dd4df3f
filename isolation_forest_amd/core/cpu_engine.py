"""CPU reference engine (numpy) — the correctness oracle for the HIP kernels.

Implements the exact algorithm semantics of the reference:

* iTree construction, Liu et al. Algorithm 2 — height limit ceil(log2(n)),
  constant-feature retry, uniform split in [min, max)
  (IsolationTree.scala:53-183),
* Extended iTree construction, Hariri et al. Alg. 2 — Gaussian hyperplane,
  L2-normalized, sparse over min(extensionLevel+1, dim) coordinates, no
  retry on degenerate splits (empty side => zero-instance leaf)
  (ExtendedIsolationTree.scala:112-270),
* path-length scoring with leaf adjustment c(numInstances)
  (IsolationTreeBase.scala:15, IsolationTree.scala:196-230),
* bagging semantics: uniform subsample without replacement (bootstrap=False,
  the composition of Binomial thinning + shuffle/slice in
  BaggedPoint.scala:114-178 + SharedTrainLogic.scala:287) or with
  replacement (bootstrap=True, Poisson bagging).

Randomness is the counter-based Philox scheme of utils/rng.py, NOT the
reference's java Random streams: we match *semantics* (the same
distributions at every decision point), not Scala RNG bit-streams — see
SURVEY.md §7 "Hard parts" #1. Self-determinism is bitwise: the same seed
yields the same forest on CPU and GPU, at any world size, because every
draw is keyed by (seed, purpose, treeId, nodeId/slot, attempt).

PRECISION CONTRACT (kept identical in ops/hip/forest_kernels.hip):
* features are float32 (core/Utils.scala:11),
* node min/max scans are float32,
* the split point / hyperplane offset is computed in float64 and then
  ROUNDED TO FLOAT32; the float32 value is what both construction-time
  partitioning and scoring compare against (the reference keeps float64 —
  we round so CPU/GPU/ONNX all agree bitwise; Avro persists the exact
  float32 value widened to double),
* EIF hyperplane dots accumulate in float32 in ascending-index order with
  un-fused multiply-add (matches the HIP kernel's __fmul_rn/__fadd_rn),
* per-row path length = int depth + float32 c(leaf count); the sum over
  trees accumulates float32 in tree order; the final score is
  2^(-mean/c(n)) in float64, cast to float32.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import numpy as np

from ..utils import det_math, rng
from ..utils.math import avg_path_length
from .forest import (
    ExtendedForest,
    Forest,
    empty_extended_forest,
    empty_forest,
)

# ---------------------------------------------------------------------------
# Bagging (reference: BaggedPoint.scala + SharedTrainLogic.scala:287)
# ---------------------------------------------------------------------------


def sample_bags(
    total_rows: int,
    num_trees: int,
    num_samples: int,
    seed: int,
    bootstrap: bool,
    tree_id_offset: int = 0,
    row_offset: int = 0,
) -> np.ndarray:
    """Sample per-tree bags of row indices, shape [num_trees, num_samples].

    ``tree_id_offset`` keys the draws by GLOBAL tree id so a tree-sharded
    multi-GPU run draws the identical forest as a single-GPU run.
    ``row_offset`` maps local row indices to global ones under row sharding
    (draws are over the local shard only; bags are iid samples of the local
    rows, statistically equivalent to the reference's partition-local
    bagging, BaggedPoint.scala:161-178).
    """
    T, n = num_trees, num_samples
    tree_ids = (np.arange(T, dtype=np.uint32) + np.uint32(tree_id_offset))[:, None]
    slots = np.arange(n, dtype=np.uint32)[None, :]
    if bootstrap:
        idx = rng.randint_below(seed, rng.P_BAG, tree_ids, slots, total_rows)
        return idx + row_offset

    if num_samples > total_rows:
        raise ValueError("cannot sample more rows than the dataset holds")

    if num_samples * 2 > total_rows:
        # dense regime: a full random permutation per tree (argsort of
        # per-(tree,row) Philox keys; ties broken stably by row id), take the
        # first n. O(T*N log N), only used when N < 2n.
        row_ids = np.arange(total_rows, dtype=np.uint32)[None, :]
        keys = rng.u32(
            seed, rng.P_BAG, tree_ids, row_ids, attempt=np.uint32(0xFFFF)
        )
        perm = np.argsort(keys, axis=1, kind="stable")
        return perm[:, :n].astype(np.int64) + row_offset

    # Sparse regime (n <= N/2): iterative collision-resolution with
    # deterministic "smallest slot wins" arbitration; expected O(1) rounds.
    idx = rng.randint_below(seed, rng.P_BAG, tree_ids, slots, total_rows)
    slot_mat = np.broadcast_to(np.arange(n, dtype=np.int64)[None, :], (T, n))
    attempt = 0
    while True:
        # per tree: among slots sharing a value, the smallest slot id wins;
        # the rest redraw with attempt+1 (deterministic, order-independent).
        order = np.lexsort((slot_mat, idx), axis=-1)  # by value, then slot
        sorted_vals = np.take_along_axis(idx, order, axis=1)
        dup = np.zeros((T, n), dtype=bool)
        dup[:, 1:] = sorted_vals[:, 1:] == sorted_vals[:, :-1]
        loser = np.zeros((T, n), dtype=bool)
        np.put_along_axis(loser, order, dup, axis=1)
        if not loser.any():
            return idx + row_offset
        attempt += 1
        if attempt > 256:
            raise RuntimeError("bag sampling failed to converge (internal bug)")
        fresh = rng.randint_below(
            seed, rng.P_BAG, tree_ids, slots, total_rows, attempt=np.uint32(attempt)
        )
        idx = np.where(loser, fresh, idx)


def feature_subsets(
    total_features: int, num_features: int, num_trees: int, seed: int,
    tree_id_offset: int = 0,
) -> np.ndarray:
    """Per-tree sorted feature subsets [num_trees, num_features] (global ids).

    Reference: shuffle(0..d-1).take(numFeatures).sorted
    (SharedTrainLogic.scala:300-304). A uniform random permutation is drawn
    as the argsort of per-(tree, feature) Philox keys.
    """
    T, d, k = num_trees, total_features, num_features
    tree_ids = (np.arange(T, dtype=np.uint32) + np.uint32(tree_id_offset))[:, None]
    feats = np.arange(d, dtype=np.uint32)[None, :]
    keys = rng.u32(seed, rng.P_FEATSUB, tree_ids, feats)
    perm = np.argsort(keys, axis=1, kind="stable")
    subset = np.sort(perm[:, :k], axis=1)
    return subset.astype(np.int32)


# ---------------------------------------------------------------------------
# Standard iTree construction (reference: IsolationTree.scala:83-183)
# ---------------------------------------------------------------------------


def _split_feature(
    bag: np.ndarray, seg: np.ndarray, features: np.ndarray, seed: int,
    tree_id: int, node_id: int, featsel_row: np.ndarray = None,
) -> Tuple[int, float, float]:
    """Draw a split feature with constant-feature retry.

    Fisher-Yates over the tree's feature subset; first feature whose values
    in this node are non-constant wins (IsolationTree.scala:124-150).
    Returns (feature, min, max) or (-1, 0, 0) when every feature is constant.
    ``featsel_row`` optionally supplies the node's precomputed Fisher-Yates
    offsets (featsel_table; bitwise-identical draws, vectorized Philox).
    """
    avail = features.copy()
    k = len(avail)
    for j in range(k):
        if featsel_row is not None:
            t = j + int(featsel_row[j])
        else:
            t = j + int(
                rng.randint_below(
                    seed, rng.P_FEATSEL, np.uint32(tree_id),
                    np.uint32(node_id), k - j, attempt=np.uint32(j),
                )
            )
        avail[j], avail[t] = avail[t], avail[j]
        f = int(avail[j])
        col = bag[seg, f]
        fmin = np.float32(col.min())
        fmax = np.float32(col.max())
        if fmin < fmax:
            return f, float(fmin), float(fmax)
    return -1, 0.0, 0.0


def draw_tables(seed: int, tree_ids: np.ndarray, max_nodes: int, k: int):
    """Vectorized per-(tree, node) Philox draws for the standard build.

    Bitwise-identical to the per-node scalar calls build_tree would
    otherwise make (same keys/purposes; rng.py is broadcast-capable) —
    one batched Philox evaluation replaces ~2*nodes scalar ones, the
    dominant cost of the CPU oracle build (profiled 80%).
    Returns (usplit [T, mn] float64, featsel [T, mn, k] int64)."""
    T = len(tree_ids)
    tg = tree_ids.astype(np.uint32)[:, None]
    ng = np.arange(max_nodes, dtype=np.uint32)[None, :]
    usplit = rng.uniform(seed, rng.P_SPLIT, tg, ng)
    featsel = np.empty((T, max_nodes, k), dtype=np.int64)
    for j in range(k):
        featsel[:, :, j] = rng.randint_below(
            seed, rng.P_FEATSEL, tg, ng, k - j, attempt=np.uint32(j))
    return usplit, featsel


def _split_value32(fmin: float, fmax: float, u: float) -> np.float32:
    """float64 split draw rounded to float32, bumped off the left edge."""
    s64 = fmin + u * (fmax - fmin)
    s32 = np.float32(s64)
    if not (np.float32(fmin) < s32):
        s32 = np.nextafter(np.float32(fmin), np.float32(np.inf), dtype=np.float32)
    return s32


def build_tree(
    bag: np.ndarray,  # [n, d_total] float32 (full-width rows; subset via indices)
    features: np.ndarray,  # sorted int32 global feature ids for this tree
    seed: int,
    tree_id: int,
    out_feature: np.ndarray,
    out_value: np.ndarray,
    out_right: np.ndarray,
    out_count: np.ndarray,
    out_value64: np.ndarray = None,
    usplit_row: np.ndarray = None,   # [max_nodes] precomputed split uniforms
    featsel_tab: np.ndarray = None,  # [max_nodes, k] precomputed FY offsets
    leaf_lut: np.ndarray = None,     # [n+1] precomputed c(m) float32
) -> int:
    """Build one iTree into pre-order SoA rows; returns node count."""
    n = bag.shape[0]
    height_limit = int(math.ceil(math.log2(max(n, 2))))
    rows = np.arange(n, dtype=np.int64)

    next_id = 0
    # stack entries: (segment_rows, height, patch_parent) — patch_parent is the
    # node whose `right` field points to the node this entry will emit (-1 root/left)
    stack = [(rows, 0, -1)]
    while stack:
        seg, height, patch = stack.pop()
        node = next_id
        next_id += 1
        if patch >= 0:
            out_right[patch] = node
        m = len(seg)
        if m <= 1 or height >= height_limit:
            out_feature[node] = Forest.LEAF
            out_value[node] = (leaf_lut[m] if leaf_lut is not None
                               else avg_path_length(m))
            out_count[node] = m
            continue
        f, fmin, fmax = _split_feature(
            bag, seg, features, seed, tree_id, node,
            featsel_tab[node] if featsel_tab is not None else None)
        if f < 0:
            out_feature[node] = Forest.LEAF
            out_value[node] = (leaf_lut[m] if leaf_lut is not None
                               else avg_path_length(m))
            out_count[node] = m
            continue
        u = (float(usplit_row[node]) if usplit_row is not None
             else float(rng.uniform(seed, rng.P_SPLIT, np.uint32(tree_id),
                                    np.uint32(node))))
        s32 = _split_value32(fmin, fmax, u)
        mask = bag[seg, f] < s32
        left = seg[mask]
        right = seg[~mask]
        out_feature[node] = f
        out_value[node] = s32
        if out_value64 is not None:
            out_value64[node] = float(s32)
        out_count[node] = -1
        # pre-order: left child = node+1 (immediately next); push right first
        stack.append((right, height + 1, node))
        stack.append((left, height + 1, -1))
    return next_id


def build_forest(
    X: np.ndarray,  # [N, d] float32
    bag_idx: np.ndarray,  # [T, n] int64
    feat_sub: np.ndarray,  # [T, k] int32
    seed: int,
    num_samples: int,
    num_features: int,
    total_num_features: int,
    tree_id_offset: int = 0,
) -> Forest:
    T, n = bag_idx.shape
    max_nodes = 2 * n - 1 if n >= 1 else 1
    forest = empty_forest(T, max_nodes, num_samples, num_features, total_num_features)
    k = feat_sub.shape[1]
    leaf_lut = avg_path_length(np.arange(n + 1)).astype(np.float32)
    # chunk the draw tables so memory stays bounded at large tree counts
    # (featsel is [chunk, max_nodes, k] int64)
    chunk = max(1, min(T, int(64e6 // max(max_nodes * k, 1)) or 1))
    for t0 in range(0, T, chunk):
        t1 = min(T, t0 + chunk)
        tree_ids = np.arange(t0, t1, dtype=np.int64) + tree_id_offset
        usplit, featsel = draw_tables(seed, tree_ids, max_nodes, k)
        for t in range(t0, t1):
            bag = X[bag_idx[t]]
            nc = build_tree(
                bag,
                feat_sub[t],
                seed,
                t + tree_id_offset,
                forest.feature[t],
                forest.value[t],
                forest.right[t],
                forest.num_instances[t],
                forest.value64[t],
                usplit_row=usplit[t - t0],
                featsel_tab=featsel[t - t0],
                leaf_lut=leaf_lut,
            )
            forest.node_count[t] = nc
    return forest


# ---------------------------------------------------------------------------
# Extended iTree construction (reference: ExtendedIsolationTree.scala:112-270)
# ---------------------------------------------------------------------------


def _gaussian(seed: int, tree_id: int, counter: int) -> float:
    u1, u2 = rng.uniform2(seed, rng.P_EIF_NORMAL, np.uint32(tree_id), np.uint32(counter))
    # Box-Muller with bitwise-deterministic log/cos (utils/det_math.py):
    # the HIP device implementation produces identical doubles.
    return float(det_math.det_gaussian(u1, u2))


def draw_tables_extended(seed: int, tree_ids: np.ndarray, max_nodes: int,
                         k: int, nnz: int):
    """Vectorized per-(tree, node, slot) Philox draws for the EIF build —
    bitwise-identical counters to the per-node calls in _draw_hyperplane
    (same purposes/keys, rng.py broadcasts). Returns
    (fy [T,mn,nnz] int64, w32 [T,mn,nnz] float32 unnormalized gaussians,
    uint [T,mn,nnz] float64 intercept uniforms)."""
    tg = tree_ids.astype(np.uint32)[:, None, None]
    ng = np.arange(max_nodes, dtype=np.int64)[None, :, None]
    jarr = np.arange(nnz, dtype=np.int64)[None, None, :]
    fy = rng.randint_below(
        seed, rng.P_EIF_COORD, tg, ng.astype(np.uint32),
        (k - jarr), attempt=jarr.astype(np.uint32),
    )
    slot = (ng * 4096 + jarr).astype(np.uint32)
    u1, u2 = rng.uniform2(seed, rng.P_EIF_NORMAL, tg, slot)
    w32 = det_math.det_gaussian(u1, u2).astype(np.float32)
    uint = rng.uniform(seed, rng.P_EIF_INTERCEPT, tg, slot).astype(np.float64)
    return fy, w32, uint


def _draw_hyperplane(
    bag: np.ndarray, seg: np.ndarray, features: np.ndarray, nnz: int,
    seed: int, tree_id: int, node_id: int, tabs=None,
):
    """Returns (idx[nnz] int32 ascending, w[nnz] float32, offset64, offset32)
    or None when the weight vector has zero norm (theoretical leaf case,
    ExtendedIsolationTree.scala:183-184)."""
    k = len(features)
    avail = features.copy()
    if tabs is not None:
        fy = tabs[0][node_id]
        w = tabs[1][node_id].copy()
    else:
        jarr = np.arange(nnz, dtype=np.uint32)
        # batched Philox draws — identical counters (and therefore bitwise-
        # identical values) to per-j scalar calls, ~5x faster per node
        fy = rng.randint_below(
            seed, rng.P_EIF_COORD, np.uint32(tree_id), np.uint32(node_id),
            (k - jarr.astype(np.int64)), attempt=jarr,
        )
    for j in range(nnz):
        t = j + int(fy[j])
        avail[j], avail[t] = avail[t], avail[j]
    coords = np.array(sorted(int(c) for c in avail[:nnz]), dtype=np.int32)

    if tabs is None:
        # weights drawn AFTER sorting, keyed by slot j over sorted coords
        base = node_id * 4096  # nnz-slot stride; nnz <= 4096 by construction
        u1, u2 = rng.uniform2(
            seed, rng.P_EIF_NORMAL, np.uint32(tree_id),
            np.uint32(base) + jarr,
        )
        w = det_math.det_gaussian(u1, u2).astype(np.float32)
    # L2 normalize in float32, sequential accumulation
    acc = np.float32(0.0)
    for j in range(nnz):
        acc = np.float32(acc + np.float32(w[j] * w[j]))
    if acc <= np.float32(0.0):
        return None
    norm = np.float32(np.sqrt(acc))
    w = (w / norm).astype(np.float32)

    # intercepts per sorted coordinate; offset in float64
    sub = bag[np.ix_(seg, coords)]
    mn = sub.min(axis=0).astype(np.float32).astype(np.float64)
    mx = sub.max(axis=0).astype(np.float32).astype(np.float64)
    if tabs is not None:
        u = tabs[2][node_id]
    else:
        u = rng.uniform(
            seed, rng.P_EIF_INTERCEPT, np.uint32(tree_id),
            np.uint32(base) + jarr,
        ).astype(np.float64)
    intercepts = mn + u * (mx - mn)
    off64 = 0.0
    for j in range(nnz):
        off64 += float(w[j]) * float(intercepts[j])
    return coords, w, off64, np.float32(off64)


def _dot32(bag: np.ndarray, seg: np.ndarray, coords: np.ndarray, w: np.ndarray):
    """Sequential float32 dot in ascending-index order (no FMA)."""
    acc = np.zeros(len(seg), dtype=np.float32)
    for j in range(len(coords)):
        prod = (w[j] * bag[seg, coords[j]]).astype(np.float32)
        acc = (acc + prod).astype(np.float32)
    return acc


def build_extended_tree(
    bag: np.ndarray,
    features: np.ndarray,
    nnz: int,
    seed: int,
    tree_id: int,
    fr: ExtendedForest,
    t: int,
    tabs=None,
    leaf_lut: np.ndarray = None,
) -> int:
    n = bag.shape[0]
    height_limit = int(math.ceil(math.log2(max(n, 2))))
    rows = np.arange(n, dtype=np.int64)
    next_id = 0
    stack = [(rows, 0, -1)]
    while stack:
        seg, height, patch = stack.pop()
        node = next_id
        next_id += 1
        if patch >= 0:
            fr.right[t, patch] = node
        m = len(seg)
        if m <= 1 or height >= height_limit:
            fr.feature[t, node] = ExtendedForest.LEAF
            fr.value[t, node] = (leaf_lut[m] if leaf_lut is not None
                                 else avg_path_length(m))
            fr.num_instances[t, node] = m
            continue
        hp = _draw_hyperplane(bag, seg, features, nnz, seed, tree_id, node,
                              tabs=tabs)
        if hp is None:
            fr.feature[t, node] = ExtendedForest.LEAF
            fr.value[t, node] = (leaf_lut[m] if leaf_lut is not None
                                 else avg_path_length(m))
            fr.num_instances[t, node] = m
            continue
        coords, w, off64, off32 = hp
        dots = _dot32(bag, seg, coords, w)
        mask = dots < off32
        left = seg[mask]
        right = seg[~mask]
        # NO retry on degenerate splits: empty side becomes a 0-instance leaf
        fr.feature[t, node] = nnz
        fr.value[t, node] = off32
        fr.offset64[t, node] = off64
        fr.hyper_idx[t, node, :nnz] = coords
        fr.hyper_w[t, node, :nnz] = w
        fr.num_instances[t, node] = -1
        stack.append((right, height + 1, node))
        stack.append((left, height + 1, -1))
    return next_id


def build_extended_forest(
    X: np.ndarray,
    bag_idx: np.ndarray,
    feat_sub: np.ndarray,
    seed: int,
    num_samples: int,
    num_features: int,
    total_num_features: int,
    extension_level: int,
    tree_id_offset: int = 0,
) -> ExtendedForest:
    T, n = bag_idx.shape
    height_limit = int(math.ceil(math.log2(max(n, 2))))
    max_nodes = 2 ** (height_limit + 1) - 1
    nnz = min(extension_level + 1, num_features)
    fr = empty_extended_forest(
        T, max_nodes, nnz, num_samples, num_features, total_num_features,
        extension_level,
    )
    k = feat_sub.shape[1]
    leaf_lut = avg_path_length(np.arange(n + 1)).astype(np.float32)
    chunk = max(1, min(T, int(32e6 // max(max_nodes * max(nnz, 1), 1)) or 1))
    for t0 in range(0, T, chunk):
        t1 = min(T, t0 + chunk)
        tree_ids = np.arange(t0, t1, dtype=np.int64) + tree_id_offset
        fy_t, w_t, u_t = draw_tables_extended(seed, tree_ids, max_nodes, k,
                                              nnz)
        for t in range(t0, t1):
            bag = X[bag_idx[t]]
            fr.node_count[t] = build_extended_tree(
                bag, feat_sub[t], nnz, seed, t + tree_id_offset, fr, t,
                tabs=(fy_t[t - t0], w_t[t - t0], u_t[t - t0]),
                leaf_lut=leaf_lut,
            )
    return fr


# ---------------------------------------------------------------------------
# Scoring (reference: IsolationTree.scala:196-230, IsolationForestModel.scala:116-151)
# ---------------------------------------------------------------------------


def path_lengths(forest: Forest, X: np.ndarray) -> np.ndarray:
    """Sum of per-tree path lengths for each row, float32 [N].

    Accumulates over trees in tree order with float32 adds, matching the
    HIP traversal kernel bit-for-bit. Dispatches to a vectorized
    fixed-trip walk (mask-free; ~6x) that is bitwise-identical to the
    masked reference walk below (tests/test_cpu_engine.py pins both
    against each other); set IFA_CPU_WALK_REF=1 to force the reference.
    """
    import os as _os

    if _os.environ.get("IFA_CPU_WALK_REF") != "1":
        return _path_lengths_fast(forest, X)
    return _path_lengths_ref(forest, X)


def _path_lengths_fast(forest: Forest, X: np.ndarray) -> np.ndarray:
    """Vectorized fixed-trip walk (the v4 kernel's structure in numpy):
    leaves self-loop, depth is folded into f32 leaf values at pack time,
    splits compare x(f32, promoted) < value64 — identical semantics and
    f32 accumulation order to _path_lengths_ref, with no per-iteration
    masking or nonzero() calls."""
    N = X.shape[0]
    T, mn = forest.feature.shape
    feat = forest.feature
    internal = feat >= 0
    leaf = feat == Forest.LEAF
    right = forest.right.astype(np.int64)
    v64_all = (forest.value64 if getattr(forest, "value64", None) is not None
               else forest.value.astype(np.float64))
    # pack: feature (sentinel column d at leaves/pads), self-loop rights,
    # -inf thresholds at leaves (the +inf sentinel column compares false),
    # depth-folded f32 leaf values
    from ..ops.gpu_engine import _node_depths

    depth = _node_depths(feat, forest.right)
    d = X.shape[1]
    ids = np.broadcast_to(np.arange(mn, dtype=np.int64)[None, :], (T, mn))
    fcol = np.where(internal, feat.astype(np.int64), d)
    nxt_r = np.where(internal, right, ids)
    thr = np.where(internal, v64_all, -np.inf)
    leafval = np.where(
        leaf, depth.astype(np.float32) + forest.value.astype(np.float32),
        np.float32(0.0)).astype(np.float32)
    max_depth = int(depth[internal | leaf].max()) if (internal | leaf).any() else 0

    Xa = np.concatenate(
        [X, np.full((N, 1), np.inf, dtype=X.dtype)], axis=1)
    rows = np.arange(N)
    total = np.zeros(N, dtype=np.float32)
    for t in range(T):
        cur = np.zeros(N, dtype=np.int64)
        f_t, r_t, thr_t = fcol[t], nxt_r[t], thr[t]
        for _ in range(max_depth):
            f = f_t[cur]
            go_left = Xa[rows, f] < thr_t[cur]
            cur = np.where(go_left, cur + 1, r_t[cur])
        total = (total + leafval[t][cur]).astype(np.float32)
    return total


def _path_lengths_ref(forest: Forest, X: np.ndarray) -> np.ndarray:
    """Masked reference walk (kept verbatim as the differential anchor)."""
    N = X.shape[0]
    total = np.zeros(N, dtype=np.float32)
    # splits compare in FLOAT64 against the exact persisted values, matching
    # the reference's Float-promoted-to-Double compare
    # (IsolationTree.scala:213-229): x < splitValue with splitValue a Double.
    # For our own forests value64 is exactly the f32 split, so this is
    # bitwise-identical to an f32 compare; for loaded foreign models it
    # removes the knife-edge flips f32 rounding produced.
    v64_all = (forest.value64 if getattr(forest, "value64", None) is not None
               else forest.value.astype(np.float64))
    for t in range(forest.num_trees):
        feat = forest.feature[t]
        val = forest.value[t]
        v64 = v64_all[t]
        right = forest.right[t]
        cur = np.zeros(N, dtype=np.int64)
        depth = np.zeros(N, dtype=np.int32)
        active = feat[cur] >= 0
        while active.any():
            f = feat[cur[active]]
            x = X[np.nonzero(active)[0], f]
            go_left = x < v64[cur[active]]
            nxt = np.where(go_left, cur[active] + 1, right[cur[active]])
            cur[active] = nxt
            depth[active] += 1
            active2 = np.zeros(N, dtype=bool)
            active2[np.nonzero(active)[0]] = feat[nxt] >= 0
            active = active2
        leaf_term = val[cur]  # precomputed c(numInstances)
        total = (total + (depth.astype(np.float32) + leaf_term)).astype(np.float32)
    return total


def path_lengths_extended(fr: ExtendedForest, X: np.ndarray) -> np.ndarray:
    N = X.shape[0]
    total = np.zeros(N, dtype=np.float32)
    nnz = fr.nnz
    for t in range(fr.num_trees):
        feat = fr.feature[t]
        val = fr.value[t]
        right = fr.right[t]
        hidx = fr.hyper_idx[t]
        hw = fr.hyper_w[t]
        cur = np.zeros(N, dtype=np.int64)
        depth = np.zeros(N, dtype=np.int32)
        active = feat[cur] >= 0
        while active.any():
            rows = np.nonzero(active)[0]
            nodes = cur[rows]
            acc = np.zeros(len(rows), dtype=np.float32)
            for j in range(nnz):
                c = hidx[nodes, j]
                wj = hw[nodes, j]
                prod = (wj * X[rows, c]).astype(np.float32)
                acc = (acc + prod).astype(np.float32)
            go_left = acc < val[nodes]
            nxt = np.where(go_left, nodes + 1, right[nodes])
            cur[rows] = nxt
            depth[rows] += 1
            active2 = np.zeros(N, dtype=bool)
            active2[rows] = feat[nxt] >= 0
            active = active2
        leaf_term = val[cur]
        total = (total + (depth.astype(np.float32) + leaf_term)).astype(np.float32)
    return total


def scores_from_path_sums(path_sum: np.ndarray, num_trees: int, num_samples: int):
    """score = 2^(-(mean path length)/c(n)), float32.

    Reference formula: IsolationForestModel.scala:135-138."""
    mean32 = (path_sum / np.float32(num_trees)).astype(np.float32)
    c = np.float32(avg_path_length(num_samples))
    return np.exp2(-(mean32.astype(np.float64) / float(c))).astype(np.float32)


def score_forest(forest: Forest, X: np.ndarray) -> np.ndarray:
    ps = path_lengths(forest, X)
    return scores_from_path_sums(ps, forest.num_trees, forest.num_samples)


def score_extended_forest(fr: ExtendedForest, X: np.ndarray) -> np.ndarray:
    ps = path_lengths_extended(fr, X)
    return scores_from_path_sums(ps, fr.num_trees, fr.num_samples)
