"""Model save/load in the reference's exact on-disk layout.

Layout (IsolationForestModelReadWrite.scala:210-323,
IsolationForestModelReadWriteUtils.scala:97-187):

    <path>/metadata/part-00000    one-line JSON: {class, timestamp,
                                  sparkVersion, uid, paramMap, ...extras}
    <path>/data/part-00000-<uuid>-c000.avro
                                  pre-order NodeData records (sentinels:
                                  leaf -> children=-1, splitAttribute=-1,
                                  splitValue=0.0; internal -> numInstances=-1)

The class strings written are the reference's
(``com.linkedin.relevance.isolationforest.IsolationForestModel`` /
``...extended.ExtendedIsolationForestModel``) so the reference's readers
and its ONNX converter accept our files unchanged; our reader accepts
both those and this package's own names.
"""

from __future__ import annotations

import glob
import json
import os
import shutil
import time
import uuid
from typing import Dict

import numpy as np

from ..core.forest import ExtendedForest, Forest, empty_extended_forest, empty_forest
from ..utils.math import avg_path_length
from ..utils.params import ExtendedParams, Params
from . import avro_io

STANDARD_MODEL_CLASS = "com.linkedin.relevance.isolationforest.IsolationForestModel"
EXTENDED_MODEL_CLASS = (
    "com.linkedin.relevance.isolationforest.extended.ExtendedIsolationForestModel"
)
STANDARD_EST_CLASS = "com.linkedin.relevance.isolationforest.IsolationForest"
EXTENDED_EST_CLASS = (
    "com.linkedin.relevance.isolationforest.extended.ExtendedIsolationForest"
)
VERSION_TAG = "isolation-forest-amd/0.1.0"


# ---------------------------------------------------------------------------
# metadata JSON
# ---------------------------------------------------------------------------


def _write_metadata(path: str, payload: Dict):
    os.makedirs(os.path.join(path, "metadata"), exist_ok=True)
    with open(os.path.join(path, "metadata", "part-00000"), "w") as f:
        f.write(json.dumps(payload, separators=(",", ":")))
    # Spark writes an (empty) _SUCCESS marker next to the text file
    open(os.path.join(path, "metadata", "_SUCCESS"), "w").close()


def _read_metadata(path: str) -> Dict:
    candidates = sorted(glob.glob(os.path.join(path, "metadata", "part-*")))
    if not candidates:
        raise FileNotFoundError(f"no metadata part file under {path}/metadata")
    with open(candidates[0]) as f:
        return json.loads(f.readline())


def _param_map_json(params: Params) -> Dict:
    # JSON types mirror Spark's Param.jsonEncode as seen in the golden
    # fixtures: DoubleParam -> number, IntParam -> int, Boolean -> bool.
    out = {}
    for name, value in params.to_dict().items():
        out[name] = value
    return out


# ---------------------------------------------------------------------------
# save
# ---------------------------------------------------------------------------


def _prep_dir(path: str, overwrite: bool):
    if os.path.exists(path):
        if not overwrite:
            raise FileExistsError(
                f"{path} already exists; use overwrite=True (write.overwrite.save)"
            )
        shutil.rmtree(path)
    os.makedirs(path, exist_ok=True)


def standard_node_records(forest: Forest):
    """Yield EnsembleNodeData rows (IsolationForestModelReadWrite.scala:82-147)."""
    for t in range(forest.num_trees):
        nc = int(forest.node_count[t])
        for i in range(nc):
            if forest.feature[t, i] == Forest.LEAF:
                node = {
                    "id": i,
                    "leftChild": -1,
                    "rightChild": -1,
                    "splitAttribute": -1,
                    "splitValue": 0.0,
                    "numInstances": int(forest.num_instances[t, i]),
                }
            else:
                v64 = (
                    float(forest.value64[t, i])
                    if forest.value64 is not None
                    else float(forest.value[t, i])
                )
                node = {
                    "id": i,
                    "leftChild": i + 1,
                    "rightChild": int(forest.right[t, i]),
                    "splitAttribute": int(forest.feature[t, i]),
                    "splitValue": v64,
                    "numInstances": -1,
                }
            yield {"treeID": t, "nodeData": node}


def extended_node_records(fr: ExtendedForest):
    for t in range(fr.num_trees):
        nc = int(fr.node_count[t])
        for i in range(nc):
            if fr.feature[t, i] == ExtendedForest.LEAF:
                node = {
                    "id": i,
                    "leftChild": -1,
                    "rightChild": -1,
                    "indices": [],
                    "weights": [],
                    "offset": 0.0,
                    "numInstances": int(fr.num_instances[t, i]),
                }
            else:
                k = int(fr.feature[t, i])
                node = {
                    "id": i,
                    "leftChild": i + 1,
                    "rightChild": int(fr.right[t, i]),
                    "indices": [int(v) for v in fr.hyper_idx[t, i, :k]],
                    "weights": [float(v) for v in fr.hyper_w[t, i, :k]],
                    "offset": float(fr.offset64[t, i]),
                    "numInstances": -1,
                }
            yield {"treeID": t, "extendedNodeData": node}


def _flat_live(forest):
    """Row-major (tree asc, id asc) selection of live node slots — the same
    record order the generators yield. Returns (tree_col, id_col, flatidx)."""
    T, mn = forest.feature.shape
    nc = forest.node_count.astype(np.int64)
    live = np.arange(mn, dtype=np.int64)[None, :] < nc[:, None]
    flat = np.nonzero(live.reshape(-1))[0]
    tree_col = flat // mn
    id_col = flat % mn
    return tree_col, id_col, flat


def _fast_save_standard(path: str, forest, codec: str):
    from . import avro_fast

    tree_col, id_col, flat = _flat_live(forest)
    feat = forest.feature.reshape(-1)[flat].astype(np.int64)
    leaf = feat == Forest.LEAF
    right = forest.right.reshape(-1)[flat].astype(np.int64)
    v64 = (forest.value64 if forest.value64 is not None else
           forest.value.astype(np.float64)).reshape(-1)[flat]
    ni = forest.num_instances.reshape(-1)[flat].astype(np.int64)
    recs = {
        "treeID": tree_col,
        "id": id_col,
        "leftChild": np.where(leaf, -1, id_col + 1),
        "rightChild": np.where(leaf, -1, right),
        "splitAttribute": np.where(leaf, -1, feat),
        "splitValue": np.where(leaf, 0.0, v64),
        "numInstances": np.where(leaf, ni, -1),
    }
    payload, reclens = avro_fast.encode_standard(recs)
    avro_io.write_container_encoded(
        path, avro_io.STANDARD_SCHEMA, payload, reclens, codec=codec,
        extra_meta={avro_fast.RECLENS_KEY: avro_fast.pack_reclens(reclens)},
    )


def _fast_save_extended(path: str, fr, codec: str):
    from . import avro_fast

    tree_col, id_col, flat = _flat_live(fr)
    feat = fr.feature.reshape(-1)[flat].astype(np.int64)
    leaf = feat == ExtendedForest.LEAF
    right = fr.right.reshape(-1)[flat].astype(np.int64)
    off = fr.offset64.reshape(-1)[flat]
    ni = fr.num_instances.reshape(-1)[flat].astype(np.int64)
    counts = np.where(leaf, 0, feat)
    nnz = fr.hyper_idx.shape[2]
    indices = fr.hyper_idx.reshape(-1, nnz)[flat].astype(np.int64)
    weights = fr.hyper_w.reshape(-1, nnz)[flat].astype(np.float32)
    recs = {
        "treeID": tree_col,
        "id": id_col,
        "leftChild": np.where(leaf, -1, id_col + 1),
        "rightChild": np.where(leaf, -1, right),
        "offset": np.where(leaf, 0.0, off),
        "numInstances": np.where(leaf, ni, -1),
    }
    payload, reclens = avro_fast.encode_extended(recs, counts, indices, weights)
    avro_io.write_container_encoded(
        path, avro_io.EXTENDED_SCHEMA, payload, reclens, codec=codec,
        extra_meta={avro_fast.RECLENS_KEY: avro_fast.pack_reclens(reclens)},
    )


def save_model(model, path: str, overwrite: bool = False, codec: str = "deflate"):
    _prep_dir(path, overwrite)
    forest = model.forest
    extended = isinstance(forest, ExtendedForest)
    meta = {
        "class": EXTENDED_MODEL_CLASS if extended else STANDARD_MODEL_CLASS,
        "timestamp": int(time.time() * 1000),
        "sparkVersion": VERSION_TAG,
        "uid": model.uid,
        "paramMap": _param_map_json(model.params),
        "outlierScoreThreshold": model.outlier_score_threshold,
        "numSamples": int(forest.num_samples),
        "numFeatures": int(forest.num_features),
        "totalNumFeatures": int(forest.total_num_features),
    }
    _write_metadata(path, meta)
    os.makedirs(os.path.join(path, "data"), exist_ok=True)
    fname = f"part-00000-{uuid.uuid4()}-c000.avro"
    fpath = os.path.join(path, "data", fname)
    if extended:
        _fast_save_extended(fpath, forest, codec)
    else:
        _fast_save_standard(fpath, forest, codec)
    open(os.path.join(path, "data", "_SUCCESS"), "w").close()


class WriteHandle:
    """``model.write.overwrite.save(path)`` parity shim
    (IsolationForestModel.scala:163)."""

    def __init__(self, model, overwrite_flag: bool = False):
        self._model = model
        self._overwrite = overwrite_flag

    @property
    def overwrite(self) -> "WriteHandle":
        return WriteHandle(self._model, True)

    def save(self, path: str):
        save_model(self._model, path, overwrite=self._overwrite)


# ---------------------------------------------------------------------------
# load
# ---------------------------------------------------------------------------


def _try_fast_columns(path: str, extended: bool):
    """Vectorized decode of all data files via the ifa.reclens metadata;
    returns a concatenated column dict or None (fall back to generic)."""
    from . import avro_fast

    files = sorted(glob.glob(os.path.join(path, "data", "*.avro")))
    if not files:
        raise FileNotFoundError(f"no avro data files under {path}/data")
    parts = []
    for f in files:
        try:
            schema, meta, payload, nrec = avro_io.read_container_raw(f)
        except Exception:
            return None
        if avro_fast.RECLENS_KEY not in meta:
            return None
        reclens = avro_fast.unpack_reclens(meta[avro_fast.RECLENS_KEY])
        if len(reclens) != nrec:
            return None
        starts = np.zeros(nrec, dtype=np.int64)
        np.cumsum(reclens[:-1], out=starts[1:])
        buf = np.frombuffer(payload, dtype=np.uint8)
        try:
            cols = (avro_fast.decode_extended(buf, starts) if extended
                    else avro_fast.decode_standard(buf, starts))
        except Exception:
            return None
        parts.append(cols)
    out = {}
    for k in parts[0]:
        arrs = [p[k] for p in parts]
        if arrs[0].ndim == 2:
            w = max(a.shape[1] for a in arrs)
            arrs = [np.pad(a, [(0, 0), (0, w - a.shape[1])]) for a in arrs]
        out[k] = np.concatenate(arrs)
    # fast path requires records grouped by tree with ascending pre-order
    # ids (our writer's order); otherwise defer to the generic loader
    tid, nid = out["treeID"], out["id"]
    if len(tid) and not (
        np.all(np.diff(tid) >= 0)
        and np.all((np.diff(nid) == 1) | (np.diff(tid) > 0))
        and np.all(nid[np.concatenate([[True], np.diff(tid) > 0])] == 0)
    ):
        return None
    return out


def _forest_from_columns(c, num_samples, num_features, total_num_features) -> Forest:
    tid = c["treeID"].astype(np.int64)
    nid = c["id"].astype(np.int64)
    T = int(tid.max()) + 1 if len(tid) else 0
    nc = np.bincount(tid, minlength=max(T, 1)).astype(np.int32)
    max_nodes = int(nc.max(initial=1))
    forest = empty_forest(T, max_nodes, num_samples, num_features,
                          total_num_features)
    if T == 0:
        return forest
    forest.node_count[:] = nc
    leaf = c["leftChild"] == -1
    internal = ~leaf
    if not np.all(c["leftChild"][internal] == nid[internal] + 1):
        bad = np.nonzero(c["leftChild"][internal] != nid[internal] + 1)[0][0]
        t, i = tid[internal][bad], nid[internal][bad]
        raise ValueError(
            f"tree {t} node {i}: leftChild {c['leftChild'][internal][bad]} "
            "breaks pre-order invariant (expected id+1)"
        )
    ti = (tid, nid)
    forest.feature[ti] = np.where(leaf, Forest.LEAF,
                                  c["splitAttribute"]).astype(np.int32)
    forest.num_instances[ti] = np.where(leaf, c["numInstances"], -1)
    forest.value[ti] = np.where(
        leaf, avg_path_length(np.maximum(c["numInstances"], 0)),
        c["splitValue"].astype(np.float32),
    ).astype(np.float32)
    forest.value64[ti] = np.where(leaf, 0.0, c["splitValue"])
    forest.right[ti] = np.where(leaf, -1, c["rightChild"]).astype(np.int32)
    return forest


def _extended_forest_from_columns(c, num_samples, num_features,
                                  total_num_features, params) -> ExtendedForest:
    tid = c["treeID"].astype(np.int64)
    nid = c["id"].astype(np.int64)
    T = int(tid.max()) + 1 if len(tid) else 0
    nc = np.bincount(tid, minlength=max(T, 1)).astype(np.int32)
    max_nodes = int(nc.max(initial=1))
    counts = c["_counts"].astype(np.int64)
    nnz = int(counts.max(initial=0))
    if params.is_set("extensionLevel"):
        ext = params.get("extensionLevel")
    else:
        ext = max(nnz - 1, 0)
    nnz = max(nnz, 1)
    fr = empty_extended_forest(
        T, max_nodes, nnz, num_samples, num_features, total_num_features, ext
    )
    if T == 0:
        return fr
    fr.node_count[:] = nc
    leaf = c["leftChild"] == -1
    if not np.all(c["leftChild"][~leaf] == nid[~leaf] + 1):
        raise ValueError("pre-order invariant broken (leftChild != id+1)")
    ti = (tid, nid)
    fr.feature[ti] = np.where(leaf, ExtendedForest.LEAF,
                              counts).astype(np.int32)
    fr.num_instances[ti] = np.where(leaf, c["numInstances"], -1)
    fr.value[ti] = np.where(
        leaf, avg_path_length(np.maximum(c["numInstances"], 0)),
        c["offset"].astype(np.float32),
    ).astype(np.float32)
    fr.offset64[ti] = np.where(leaf, 0.0, c["offset"])
    fr.right[ti] = np.where(leaf, -1, c["rightChild"]).astype(np.int32)
    w = c["_indices"].shape[1]
    fr.hyper_idx[tid, nid, :w] = c["_indices"].astype(np.int32)
    fr.hyper_w[tid, nid, :w] = c["_weights"].astype(np.float32)
    return fr


def _records_by_tree(path: str, field: str):
    files = sorted(glob.glob(os.path.join(path, "data", "*.avro")))
    if not files:
        raise FileNotFoundError(f"no avro data files under {path}/data")
    trees: Dict[int, list] = {}
    for f in files:
        _, records = avro_io.read_container(f)
        for rec in records:
            trees.setdefault(rec["treeID"], []).append(rec[field])
    for t in trees:
        trees[t].sort(key=lambda n: n["id"])
    return trees


def load_model(path: str, expect_extended: bool):
    from ..models.extended_isolation_forest import ExtendedIsolationForestModel
    from ..models.isolation_forest import IsolationForestModel

    meta = _read_metadata(path)
    cls = meta.get("class", "")
    is_extended = "Extended" in cls.rsplit(".", 1)[-1]
    if expect_extended != is_extended:
        raise ValueError(
            f"model at {path} has class {cls!r}; wrong loader "
            f"(expected {'extended' if expect_extended else 'standard'})"
        )
    params_cls = ExtendedParams if is_extended else Params
    params = params_cls()
    params.apply_map(meta.get("paramMap", {}))

    num_samples = int(meta["numSamples"])
    num_features = int(meta["numFeatures"])
    total_num_features = int(meta.get("totalNumFeatures", -1))
    threshold = float(meta.get("outlierScoreThreshold", -1.0))

    cols = _try_fast_columns(path, extended=is_extended)
    if is_extended:
        if cols is not None:
            forest = _extended_forest_from_columns(
                cols, num_samples, num_features, total_num_features, params
            )
        else:
            trees = _records_by_tree(path, "extendedNodeData")
            forest = _extended_forest_from_records(
                trees, num_samples, num_features, total_num_features, params
            )
        model = ExtendedIsolationForestModel(
            uid=meta.get("uid", "extended-isolation-forest_loaded"),
            forest=forest,
            params=params,
        )
    else:
        if cols is not None:
            forest = _forest_from_columns(
                cols, num_samples, num_features, total_num_features
            )
        else:
            trees = _records_by_tree(path, "nodeData")
            forest = _forest_from_records(
                trees, num_samples, num_features, total_num_features
            )
        model = IsolationForestModel(
            uid=meta.get("uid", "isolation-forest_loaded"),
            forest=forest,
            params=params,
        )
    model.set_outlier_score_threshold(threshold)
    return model


def _forest_from_records(trees, num_samples, num_features, total_num_features) -> Forest:
    T = (max(trees) + 1) if trees else 0
    max_nodes = max((len(v) for v in trees.values()), default=1)
    forest = empty_forest(T, max_nodes, num_samples, num_features, total_num_features)
    for t, nodes in trees.items():
        forest.node_count[t] = len(nodes)
        for n in nodes:
            i = n["id"]
            if n["leftChild"] == -1:
                forest.feature[t, i] = Forest.LEAF
                forest.num_instances[t, i] = n["numInstances"]
                forest.value[t, i] = avg_path_length(n["numInstances"])
            else:
                if n["leftChild"] != i + 1:
                    raise ValueError(
                        f"tree {t} node {i}: leftChild {n['leftChild']} breaks "
                        "pre-order invariant (expected id+1)"
                    )
                forest.feature[t, i] = n["splitAttribute"]
                forest.value[t, i] = np.float32(n["splitValue"])
                forest.value64[t, i] = n["splitValue"]
                forest.right[t, i] = n["rightChild"]
                forest.num_instances[t, i] = -1
    return forest


def _extended_forest_from_records(
    trees, num_samples, num_features, total_num_features, params
) -> ExtendedForest:
    T = (max(trees) + 1) if trees else 0
    max_nodes = max((len(v) for v in trees.values()), default=1)
    nnz = 0
    for nodes in trees.values():
        for n in nodes:
            if n["indices"]:
                nnz = max(nnz, len(n["indices"]))
    if params.is_set("extensionLevel"):
        ext = params.get("extensionLevel")
    else:
        ext = max(nnz - 1, 0)
    nnz = max(nnz, 1)
    fr = empty_extended_forest(
        T, max_nodes, nnz, num_samples, num_features, total_num_features, ext
    )
    for t, nodes in trees.items():
        fr.node_count[t] = len(nodes)
        for n in nodes:
            i = n["id"]
            if n["leftChild"] == -1:
                fr.feature[t, i] = ExtendedForest.LEAF
                fr.num_instances[t, i] = n["numInstances"]
                fr.value[t, i] = avg_path_length(n["numInstances"])
            else:
                k = len(n["indices"])
                fr.feature[t, i] = k
                fr.hyper_idx[t, i, :k] = n["indices"]
                fr.hyper_w[t, i, :k] = np.asarray(n["weights"], dtype=np.float32)
                fr.offset64[t, i] = n["offset"]
                fr.value[t, i] = np.float32(n["offset"])
                fr.right[t, i] = n["rightChild"]
                fr.num_instances[t, i] = -1
    return fr


# ---------------------------------------------------------------------------
# estimator save/load (metadata only — IsolationForest.scala:114-125)
# ---------------------------------------------------------------------------


def save_estimator(estimator, path: str, overwrite: bool = False):
    from ..models.extended_isolation_forest import ExtendedIsolationForest

    _prep_dir(path, overwrite)
    extended = isinstance(estimator, ExtendedIsolationForest)
    _write_metadata(
        path,
        {
            "class": EXTENDED_EST_CLASS if extended else STANDARD_EST_CLASS,
            "timestamp": int(time.time() * 1000),
            "sparkVersion": VERSION_TAG,
            "uid": estimator.uid,
            "paramMap": _param_map_json(estimator.params),
        },
    )


def load_estimator(cls, path: str):
    meta = _read_metadata(path)
    est = cls(uid=meta.get("uid"))
    est.params.apply_map(meta.get("paramMap", {}))
    return est
