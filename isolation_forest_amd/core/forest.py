"""Flat SoA forest representation — the engine's central data structure.

The reference stores trees as linked case-class nodes (Nodes.scala:25-66)
and flattens them to pre-order NodeData records only for persistence
(IsolationForestModelReadWrite.scala:60-133). On MI355X the pre-order flat
array IS the runtime format: the scoring kernel walks it directly from
LDS/L2 and persistence is a reinterpretation, not a conversion.

Standard forest layout (padded SoA, [num_trees, max_nodes]):

* ``feature`` int32  — split attribute (global feature index) for internal
  nodes; ``-1`` for leaves; ``-2`` for padding beyond ``node_count[t]``.
* ``value``  float32 — split threshold for internal nodes; precomputed
  ``avg_path_length(num_instances)`` for leaves (so the traversal kernel
  never computes logs — reference computes it per visit,
  IsolationTree.scala:218).
* ``right``  int32  — pre-order index of the right child (within-tree);
  the LEFT child of node i is always i+1 in pre-order, so it is implicit.
* ``num_instances`` int64 — leaf row count (internal nodes: -1); kept for
  persistence parity (NodeData sentinels,
  core/IsolationForestModelReadWriteUtils.scala:30-31).

Pre-order node ids match the reference's NodeData.build ordering
(IsolationForestModelReadWrite.scala:82-133) so Avro files round-trip
against the reference's readers and the ONNX converter.

Extended forest adds per-node sparse hyperplanes with a FIXED number of
non-zero coordinates ``nnz = min(extensionLevel+1, num_features)``
(ExtendedIsolationTree.scala:155-157): ``hyper_idx`` int32 and ``hyper_w``
float32 of shape [num_trees, max_nodes, nnz] (leaf rows zero-filled), and
``value`` holds the split offset for internal nodes.
"""

from __future__ import annotations

import dataclasses
from typing import Optional

import numpy as np

from ..utils.math import avg_path_length


@dataclasses.dataclass
class Forest:
    feature: np.ndarray  # int32  [T, max_nodes]
    value: np.ndarray  # float32 [T, max_nodes]
    right: np.ndarray  # int32  [T, max_nodes]
    num_instances: np.ndarray  # int64 [T, max_nodes]
    node_count: np.ndarray  # int32 [T]
    num_samples: int  # rows per tree the forest was trained on
    num_features: int  # resolved per-tree feature-subspace size
    total_num_features: int  # input dimensionality (-1 = unknown/legacy)
    value64: np.ndarray = None  # float64 [T, max_nodes] — exact split values
    # (equal to float64(value) for freshly trained forests; preserves the
    # persisted double exactly on load so save(load(x)) == x)

    PAD = -2
    LEAF = -1

    @property
    def num_trees(self) -> int:
        return int(self.feature.shape[0])

    @property
    def max_nodes(self) -> int:
        return int(self.feature.shape[1])

    def tree_to_string(self, t: int) -> str:
        """Canonical structural fingerprint, byte-compatible with the
        reference's Node.toString (Nodes.scala:63-65) so golden-structure
        files compare EXACTLY across implementations."""
        from ..utils.javafmt import java_double

        v64 = self.value64 if self.value64 is not None else self.value.astype(np.float64)

        def rec(i: int) -> str:
            if self.feature[t, i] == Forest.LEAF:
                return f"ExternalNode(numInstances = {int(self.num_instances[t, i])})"
            left = rec(i + 1)
            right = rec(int(self.right[t, i]))
            return (
                f"InternalNode(splitAttribute = {int(self.feature[t, i])}, "
                f"splitValue = {java_double(v64[t, i])}, "
                f"leftChild = ({left}), rightChild = ({right}))"
            )

        return rec(0)

    def subtree_depth(self, t: int, i: int = 0) -> int:
        """Depth as the reference defines it (Nodes.scala:61): leaves are 0."""
        if self.feature[t, i] == Forest.LEAF:
            return 0
        return 1 + max(
            self.subtree_depth(t, i + 1), self.subtree_depth(t, int(self.right[t, i]))
        )

    def leaf_values_from_counts(self):
        """(Re)compute leaf path-length terms from num_instances in place."""
        leaf_mask = self.feature == Forest.LEAF
        counts = np.where(leaf_mask, self.num_instances, 0)
        self.value = np.where(
            leaf_mask, avg_path_length(counts), self.value
        ).astype(np.float32)

    def validate(self):
        T = self.num_trees
        assert self.value.shape == (T, self.max_nodes)
        assert self.right.shape == (T, self.max_nodes)
        assert self.num_instances.shape == (T, self.max_nodes)
        assert self.node_count.shape == (T,)
        for t in range(T):
            nc = int(self.node_count[t])
            assert nc >= 1
            assert np.all(self.feature[t, nc:] == Forest.PAD)


@dataclasses.dataclass
class ExtendedForest:
    feature: np.ndarray  # int32 [T, max_nodes]: nnz count for internal, -1 leaf, -2 pad
    value: np.ndarray  # float32 [T, max_nodes]: offset (internal) / leaf term
    right: np.ndarray  # int32 [T, max_nodes]
    num_instances: np.ndarray  # int64 [T, max_nodes] (EIF leaves may be 0)
    node_count: np.ndarray  # int32 [T]
    hyper_idx: np.ndarray  # int32 [T, max_nodes, nnz] ascending global indices
    hyper_w: np.ndarray  # float32 [T, max_nodes, nnz]
    offset64: np.ndarray  # float64 [T, max_nodes] — exact split offsets for persistence
    num_samples: int
    num_features: int
    total_num_features: int
    extension_level: int

    PAD = -2
    LEAF = -1

    @property
    def num_trees(self) -> int:
        return int(self.feature.shape[0])

    @property
    def max_nodes(self) -> int:
        return int(self.feature.shape[1])

    @property
    def nnz(self) -> int:
        return int(self.hyper_idx.shape[2])

    def tree_to_string(self, t: int) -> str:
        """Byte-compatible with the reference's ExtendedNodes toString
        (expectedExtendedTreeStructure.txt format)."""
        from ..utils.javafmt import java_double, java_float

        def rec(i: int) -> str:
            if self.feature[t, i] == ExtendedForest.LEAF:
                return f"ExtendedExternalNode(numInstances = {int(self.num_instances[t, i])})"
            k = int(self.feature[t, i])
            idx = ", ".join(str(int(v)) for v in self.hyper_idx[t, i, :k])
            w = ", ".join(java_float(v) for v in self.hyper_w[t, i, :k])
            return (
                f"ExtendedInternalNode(splitHyperplane = SplitHyperplane("
                f"indices = ({idx}), weights = ({w}), "
                f"offset = {java_double(self.offset64[t, i])}), "
                f"leftChild = ({rec(i + 1)}), "
                f"rightChild = ({rec(int(self.right[t, i]))}))"
            )

        return rec(0)

    def subtree_depth(self, t: int, i: int = 0) -> int:
        if self.feature[t, i] == ExtendedForest.LEAF:
            return 0
        return 1 + max(
            self.subtree_depth(t, i + 1), self.subtree_depth(t, int(self.right[t, i]))
        )

    def leaf_values_from_counts(self):
        leaf_mask = self.feature == ExtendedForest.LEAF
        counts = np.where(leaf_mask, self.num_instances, 0)
        self.value = np.where(
            leaf_mask, avg_path_length(counts), self.value
        ).astype(np.float32)


def empty_forest(
    num_trees: int,
    max_nodes: int,
    num_samples: int,
    num_features: int,
    total_num_features: int,
) -> Forest:
    return Forest(
        feature=np.full((num_trees, max_nodes), Forest.PAD, dtype=np.int32),
        value=np.zeros((num_trees, max_nodes), dtype=np.float32),
        right=np.full((num_trees, max_nodes), -1, dtype=np.int32),
        num_instances=np.full((num_trees, max_nodes), -1, dtype=np.int64),
        node_count=np.zeros(num_trees, dtype=np.int32),
        num_samples=num_samples,
        num_features=num_features,
        total_num_features=total_num_features,
        value64=np.zeros((num_trees, max_nodes), dtype=np.float64),
    )


def empty_extended_forest(
    num_trees: int,
    max_nodes: int,
    nnz: int,
    num_samples: int,
    num_features: int,
    total_num_features: int,
    extension_level: int,
) -> ExtendedForest:
    return ExtendedForest(
        feature=np.full((num_trees, max_nodes), ExtendedForest.PAD, dtype=np.int32),
        value=np.zeros((num_trees, max_nodes), dtype=np.float32),
        right=np.full((num_trees, max_nodes), -1, dtype=np.int32),
        num_instances=np.full((num_trees, max_nodes), -1, dtype=np.int64),
        node_count=np.zeros(num_trees, dtype=np.int32),
        hyper_idx=np.zeros((num_trees, max_nodes, nnz), dtype=np.int32),
        hyper_w=np.zeros((num_trees, max_nodes, nnz), dtype=np.float32),
        offset64=np.zeros((num_trees, max_nodes), dtype=np.float64),
        num_samples=num_samples,
        num_features=num_features,
        total_num_features=total_num_features,
        extension_level=extension_level,
    )
