"""Self-contained ONNX evaluator for the Isolation Forest export graph.

onnxruntime is not available in this offline environment, so parity tests
(and users who want portable inference without an ONNX runtime) execute the
serialized model directly: the evaluator PARSES THE BYTES the exporter wrote
(validating the wire format end-to-end) and interprets the ops the graph
uses: TreeEnsembleRegressor (ai.onnx.ml, BRANCH_LT / LEAF, AVERAGE),
Constant, Div, Neg, Pow, Less, Not, Cast.

Mirrors onnxruntime semantics for this op set: float32 tensor math, tree
threshold compare in float64 (nodes_values attribute is f32 here; compare
value promoted), aggregate AVERAGE = mean of per-tree leaf weights.
"""

from __future__ import annotations

from typing import Dict

import numpy as np

from . import protowire as pw


def _parse_attr(buf: bytes):
    f = pw.parse_message(buf)
    name = f[1][0][1].decode()
    atype = f[20][0][1] if 20 in f else None
    if 2 in f:
        return name, np.frombuffer(f[2][0][1], dtype="<f4")[0]
    if 3 in f:
        return name, pw.as_int64(f[3][0][1])
    if 4 in f:
        return name, f[4][0][1]
    if 5 in f:
        return name, _parse_tensor(f[5][0][1])
    if 7 in f:
        return name, pw.unpack_floats(f[7][0][1])
    if 8 in f:
        return name, pw.unpack_varints(f[8][0][1])
    if 9 in f:
        return name, [v.decode() for _, v in f[9]]
    raise ValueError(f"attribute {name!r} (type {atype}) has no payload")


def _parse_tensor(buf: bytes) -> np.ndarray:
    f = pw.parse_message(buf)
    dims = [v for _, v in f.get(1, [])]
    dtype = f[2][0][1] if 2 in f else 1
    if 4 in f:
        data = np.array(pw.unpack_floats(f[4][0][1]), dtype=np.float32)
    elif 5 in f:
        data = np.array(pw.unpack_varints(f[5][0][1]), dtype=np.int32)
    elif 9 in f:
        raw = f[9][0][1]
        data = np.frombuffer(raw, dtype="<f4" if dtype == 1 else "<i4")
    else:
        data = np.array([], dtype=np.float32)
    return data.reshape(dims) if dims else (data[0] if data.size else data)


def _parse_node(buf: bytes) -> Dict:
    f = pw.parse_message(buf)
    return {
        "inputs": [v.decode() for _, v in f.get(1, [])],
        "outputs": [v.decode() for _, v in f.get(2, [])],
        "op_type": f[4][0][1].decode(),
        "attrs": dict(_parse_attr(v) for _, v in f.get(5, [])),
    }


def parse_model(model_bytes: bytes) -> Dict:
    """Returns {ir_version, opsets: [(domain, version)], nodes: [...],
    inputs/outputs: [names]}."""
    f = pw.parse_message(model_bytes)
    graph = pw.parse_message(f[7][0][1])
    opsets = []
    for _, raw in f.get(8, []):
        op = pw.parse_message(raw)
        dom = op[1][0][1].decode() if 1 in op else ""
        opsets.append((dom, op[2][0][1]))
    def names(entries):
        return [pw.parse_message(raw)[1][0][1].decode() for _, raw in entries]
    return {
        "ir_version": f[1][0][1],
        "opsets": opsets,
        "nodes": [_parse_node(raw) for _, raw in graph.get(1, [])],
        "inputs": names(graph.get(11, [])),
        "outputs": names(graph.get(12, [])),
    }


def _run_tree_ensemble(attrs: Dict, X: np.ndarray) -> np.ndarray:
    tree_ids = np.asarray(attrs["nodes_treeids"], dtype=np.int64)
    node_ids = np.asarray(attrs["nodes_nodeids"], dtype=np.int64)
    feats = np.asarray(attrs["nodes_featureids"], dtype=np.int64)
    modes = attrs["nodes_modes"]
    values = np.asarray(attrs["nodes_values"], dtype=np.float32)
    true_ids = np.asarray(attrs["nodes_truenodeids"], dtype=np.int64)
    false_ids = np.asarray(attrs["nodes_falsenodeids"], dtype=np.int64)
    trees = sorted(set(int(t) for t in tree_ids))
    # per (tree, node_id) -> flat record index
    index = {(int(t), int(n)): i for i, (t, n) in enumerate(zip(tree_ids,
                                                                node_ids))}
    weights = {(int(t), int(n)): w for t, n, w in zip(
        attrs["target_treeids"], attrs["target_nodeids"],
        attrs["target_weights"])}
    is_leaf = np.array([m == "LEAF" for m in modes])

    N = X.shape[0]
    total = np.zeros(N, dtype=np.float64)
    for t in trees:
        cur = np.full(N, index[(t, 0)], dtype=np.int64)
        active = ~is_leaf[cur]
        while active.any():
            rows = np.nonzero(active)[0]
            c = cur[rows]
            go_true = X[rows, feats[c]] < values[c]
            nxt_node = np.where(go_true, true_ids[c], false_ids[c])
            cur[rows] = [index[(t, int(n))] for n in nxt_node]
            active[rows] = ~is_leaf[cur[rows]]
        leaf_w = np.array(
            [weights[(t, int(node_ids[ci]))] for ci in cur], dtype=np.float32)
        total += leaf_w
    return (total / np.float32(len(trees))).astype(np.float32)


def run(model_bytes: bytes, features: np.ndarray) -> Dict[str, np.ndarray]:
    """Execute the exported graph on float32 features [N, d]."""
    model = parse_model(model_bytes)
    X = np.ascontiguousarray(features, dtype=np.float32)
    env: Dict[str, np.ndarray] = {"features": X}
    for node in model["nodes"]:
        op = node["op_type"]
        ins = [env[i] for i in node["inputs"]]
        if op == "TreeEnsembleRegressor":
            out = _run_tree_ensemble(node["attrs"], X).reshape(-1, 1)
        elif op == "Constant":
            out = np.asarray(node["attrs"]["value"], dtype=np.float32)
        elif op == "Div":
            out = (ins[0] / ins[1]).astype(np.float32)
        elif op == "Neg":
            out = (-ins[0]).astype(np.float32)
        elif op == "Pow":
            out = np.power(ins[0], ins[1]).astype(np.float32)
        elif op == "Less":
            out = ins[0] < ins[1]
        elif op == "Not":
            out = ~ins[0]
        elif op == "Cast":
            to = int(node["attrs"]["to"])
            out = ins[0].astype(np.int32 if to == 6 else np.float32)
        else:
            raise ValueError(f"evaluator does not support op {op}")
        env[node["outputs"][0]] = out
    return {name: env[name] for name in model["outputs"]}
