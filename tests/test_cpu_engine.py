"""CPU oracle engine tests: bagging, tree invariants, scoring, determinism.

Reference test models: IsolationTreeTest.scala, BaggedPointTest.scala,
ExtendedIsolationTreeTest.scala.
"""

import math

import numpy as np
import pytest

from isolation_forest_amd.core import cpu_engine
from isolation_forest_amd.core.forest import ExtendedForest, Forest


def make_data(n=2000, d=6, seed=3):
    rs = np.random.RandomState(seed)
    return rs.normal(size=(n, d)).astype(np.float32)


class TestBagging:
    def test_without_replacement_unique(self):
        idx = cpu_engine.sample_bags(1000, 50, 256, seed=1, bootstrap=False)
        assert idx.shape == (50, 256)
        for t in range(50):
            assert len(np.unique(idx[t])) == 256
        assert idx.min() >= 0 and idx.max() < 1000

    def test_small_pool(self):
        # n == N: must be a permutation
        idx = cpu_engine.sample_bags(256, 20, 256, seed=1, bootstrap=False)
        for t in range(20):
            assert sorted(idx[t]) == list(range(256))

    def test_with_replacement(self):
        idx = cpu_engine.sample_bags(1000, 50, 256, seed=1, bootstrap=True)
        assert idx.shape == (50, 256)
        assert idx.min() >= 0 and idx.max() < 1000

    def test_determinism_and_tree_offset(self):
        a = cpu_engine.sample_bags(5000, 10, 64, seed=2, bootstrap=False)
        b = cpu_engine.sample_bags(5000, 10, 64, seed=2, bootstrap=False)
        assert np.array_equal(a, b)
        # global-tree-id keying: trees [5,10) of a 10-tree run == trees of a
        # sharded run starting at offset 5
        c = cpu_engine.sample_bags(5000, 5, 64, seed=2, bootstrap=False, tree_id_offset=5)
        assert np.array_equal(a[5:], c)

    def test_uniformity(self):
        # every row should be roughly equally likely across many trees
        idx = cpu_engine.sample_bags(500, 400, 250, seed=3, bootstrap=False)
        counts = np.bincount(idx.ravel(), minlength=500)
        expect = 400 * 250 / 500
        assert abs(counts.mean() - expect) < 1e-9
        assert counts.std() < 0.15 * expect

    def test_seed_changes_bags(self):
        a = cpu_engine.sample_bags(5000, 10, 64, seed=2, bootstrap=False)
        b = cpu_engine.sample_bags(5000, 10, 64, seed=3, bootstrap=False)
        assert not np.array_equal(a, b)


class TestFeatureSubsets:
    def test_full(self):
        fs = cpu_engine.feature_subsets(10, 10, 5, seed=1)
        for t in range(5):
            assert list(fs[t]) == list(range(10))

    def test_subset_sorted_distinct(self):
        fs = cpu_engine.feature_subsets(20, 7, 100, seed=1)
        assert fs.shape == (100, 7)
        for t in range(100):
            row = list(fs[t])
            assert row == sorted(row)
            assert len(set(row)) == 7
        # all features get picked somewhere
        assert len(np.unique(fs)) == 20

    def test_offset_consistency(self):
        a = cpu_engine.feature_subsets(20, 7, 10, seed=1)
        b = cpu_engine.feature_subsets(20, 7, 5, seed=1, tree_id_offset=5)
        assert np.array_equal(a[5:], b)


def check_tree_invariants(forest, t, num_samples):
    """Pre-order structure, instance conservation, height limit."""
    nc = int(forest.node_count[t])
    height_limit = math.ceil(math.log2(max(num_samples, 2)))
    leaf_sum = 0
    stack = [(0, 0)]
    visited = 0
    order = []
    while stack:
        i, h = stack.pop()
        order.append(i)
        visited += 1
        assert h <= height_limit
        if forest.feature[t, i] == forest.LEAF:
            leaf_sum += int(forest.num_instances[t, i])
        else:
            r = int(forest.right[t, i])
            assert 0 < r < nc
            stack.append((r, h + 1))
            stack.append((i + 1, h + 1))
    assert visited == nc
    if isinstance(forest, Forest):
        # standard IF: every leaf >= 1 instance => nodes <= 2n-1
        assert nc <= 2 * num_samples - 1
    assert leaf_sum == num_samples
    # pre-order ids are exactly 0..nc-1 with left = parent+1: the DFS above
    # (right pushed first) must visit ids in descending-stack pre-order
    assert sorted(order) == list(range(nc))


class TestStandardBuild:
    def test_invariants_and_determinism(self):
        X = make_data()
        bag = cpu_engine.sample_bags(len(X), 8, 256, seed=1, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 8, seed=1)
        f1 = cpu_engine.build_forest(X, bag, fs, 1, 256, 6, 6)
        f2 = cpu_engine.build_forest(X, bag, fs, 1, 256, 6, 6)
        for t in range(8):
            check_tree_invariants(f1, t, 256)
            assert f1.tree_to_string(t) == f2.tree_to_string(t)

    def test_split_attributes_within_subset(self):
        X = make_data(d=10)
        bag = cpu_engine.sample_bags(len(X), 6, 128, seed=1, bootstrap=False)
        fs = cpu_engine.feature_subsets(10, 3, 6, seed=1)
        f = cpu_engine.build_forest(X, bag, fs, 1, 128, 3, 10)
        for t in range(6):
            feats = f.feature[t, : f.node_count[t]]
            internal = feats[feats >= 0]
            assert set(internal.tolist()) <= set(fs[t].tolist())

    def test_constant_features_root_leaf(self):
        # all-identical rows: no splittable feature => root is a leaf
        # (reference: IsolationForestModelWriteReadTest:186-249)
        X = np.ones((500, 4), dtype=np.float32)
        bag = cpu_engine.sample_bags(500, 3, 64, seed=1, bootstrap=False)
        fs = cpu_engine.feature_subsets(4, 4, 3, seed=1)
        f = cpu_engine.build_forest(X, bag, fs, 1, 64, 4, 4)
        for t in range(3):
            assert f.node_count[t] == 1
            assert f.feature[t, 0] == Forest.LEAF
            assert f.num_instances[t, 0] == 64

    def test_partial_constant_feature_retry(self):
        # one constant column among informative ones: must never be chosen
        X = make_data(d=4)
        X[:, 2] = 5.0
        bag = cpu_engine.sample_bags(len(X), 5, 128, seed=1, bootstrap=False)
        fs = cpu_engine.feature_subsets(4, 4, 5, seed=1)
        f = cpu_engine.build_forest(X, bag, fs, 1, 128, 4, 4)
        for t in range(5):
            feats = f.feature[t, : f.node_count[t]]
            assert 2 not in feats[feats >= 0]

    def test_deep_tree_on_full_data(self):
        # analog of IsolationTreeTest.generateIsolationTreeTest: with enough
        # rows the tree reaches its height limit
        X = make_data(n=4096, d=6)
        bag = np.arange(4096, dtype=np.int64)[None, :]
        fs = cpu_engine.feature_subsets(6, 6, 1, seed=1)
        f = cpu_engine.build_forest(X, bag, fs, 1, 4096, 6, 6)
        assert f.subtree_depth(0) == 12  # ceil(log2(4096))


class TestScoring:
    def test_handbuilt_tree_path_lengths(self):
        # the reference's exact hand-built-tree goldens
        # (IsolationTreeTest.pathLengthTest)
        import isolation_forest_amd.core.forest as fmod

        f = fmod.empty_forest(1, 3, num_samples=30, num_features=1, total_num_features=1)
        f.feature[0, 0] = 0
        f.value[0, 0] = 1.5
        f.value64[0, 0] = 1.5
        f.right[0, 0] = 2
        f.node_count[0] = 3
        f.feature[0, 1] = Forest.LEAF
        f.num_instances[0, 1] = 10
        f.feature[0, 2] = Forest.LEAF
        f.num_instances[0, 2] = 20
        f.leaf_values_from_counts()
        ps = cpu_engine.path_lengths(f, np.array([[1.0], [2.0]], dtype=np.float32))
        assert ps[0] == np.float32(4.7488804)
        assert ps[1] == np.float32(6.143309)

    def test_scores_in_unit_interval(self):
        X = make_data()
        bag = cpu_engine.sample_bags(len(X), 16, 256, seed=1, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 16, seed=1)
        f = cpu_engine.build_forest(X, bag, fs, 1, 256, 6, 6)
        s = cpu_engine.score_forest(f, X)
        assert s.dtype == np.float32
        assert s.min() > 0.0 and s.max() < 1.0

    def test_outliers_score_higher(self, gaussian_data):
        X, y = gaussian_data
        bag = cpu_engine.sample_bags(len(X), 50, 256, seed=1, bootstrap=False)
        fs = cpu_engine.feature_subsets(X.shape[1], X.shape[1], 50, seed=1)
        f = cpu_engine.build_forest(X, bag, fs, 1, 256, X.shape[1], X.shape[1])
        s = cpu_engine.score_forest(f, X)
        assert s[y == 1].mean() > s[y == 0].mean() + 0.1


class TestExtendedBuild:
    def test_invariants_and_determinism(self):
        X = make_data()
        bag = cpu_engine.sample_bags(len(X), 6, 256, seed=1, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 6, seed=1)
        f1 = cpu_engine.build_extended_forest(X, bag, fs, 1, 256, 6, 6, 5)
        f2 = cpu_engine.build_extended_forest(X, bag, fs, 1, 256, 6, 6, 5)
        assert isinstance(f1, ExtendedForest)
        for t in range(6):
            assert f1.tree_to_string(t) == f2.tree_to_string(t)
            nc = int(f1.node_count[t])
            internal = f1.feature[t, :nc] >= 0
            assert np.all(f1.feature[t, :nc][internal] == 6)  # nnz = ext+1 = 6

    def test_hyperplanes_normalized_sorted(self):
        X = make_data()
        bag = cpu_engine.sample_bags(len(X), 4, 128, seed=2, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 4, seed=2)
        f = cpu_engine.build_extended_forest(X, bag, fs, 2, 128, 6, 6, 3)
        for t in range(4):
            nc = int(f.node_count[t])
            for i in range(nc):
                k = int(f.feature[t, i])
                if k < 0:
                    continue
                assert k == 4  # min(3+1, 6)
                idx = f.hyper_idx[t, i, :k]
                assert list(idx) == sorted(set(int(v) for v in idx))
                norm = float(np.sqrt(np.sum(f.hyper_w[t, i, :k].astype(np.float64) ** 2)))
                assert norm == pytest.approx(1.0, abs=1e-5)

    def test_extension_level_zero_single_coord(self):
        # ext=0 => exactly 1 non-zero coordinate (axis-parallel EIF)
        # (ExtendedIsolationTreeTest:197-239)
        X = make_data()
        bag = cpu_engine.sample_bags(len(X), 4, 128, seed=2, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 4, seed=2)
        f = cpu_engine.build_extended_forest(X, bag, fs, 2, 128, 6, 6, 0)
        assert f.nnz == 1
        for t in range(4):
            nc = int(f.node_count[t])
            internal = f.feature[t, :nc] >= 0
            assert np.all(f.feature[t, :nc][internal] == 1)

    def test_zero_size_leaves_allowed(self):
        # EIF has no degenerate-split retry: zero-instance leaves can occur
        # and contribute avg_path_length(0) == 0
        X = make_data(n=3000)
        bag = cpu_engine.sample_bags(len(X), 20, 256, seed=5, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 20, seed=5)
        f = cpu_engine.build_extended_forest(X, bag, fs, 5, 256, 6, 6, 5)
        zero_leaves = 0
        for t in range(20):
            nc = int(f.node_count[t])
            for i in range(nc):
                if f.feature[t, i] == ExtendedForest.LEAF:
                    n = int(f.num_instances[t, i])
                    assert n >= 0
                    if n == 0:
                        zero_leaves += 1
                        assert f.value[t, i] == 0.0
        # sums conserved even with zero leaves
        for t in range(20):
            nc = int(f.node_count[t])
            leaf = f.feature[t, :nc] == ExtendedForest.LEAF
            assert f.num_instances[t, :nc][leaf].sum() == 256

    def test_scores_sane(self, gaussian_data):
        X, y = gaussian_data
        bag = cpu_engine.sample_bags(len(X), 40, 256, seed=1, bootstrap=False)
        fs = cpu_engine.feature_subsets(X.shape[1], X.shape[1], 40, seed=1)
        f = cpu_engine.build_extended_forest(
            X, bag, fs, 1, 256, X.shape[1], X.shape[1], X.shape[1] - 1
        )
        s = cpu_engine.score_extended_forest(f, X)
        assert s.min() > 0.0 and s.max() < 1.0
        assert s[y == 1].mean() > s[y == 0].mean() + 0.1


class TestFastWalkEquivalence:
    """_path_lengths_fast (vectorized fixed-trip walk) must be bitwise-
    identical to the masked reference walk, including foreign f64
    knife-edge splits and NaN rows."""

    def test_bitwise_on_random_and_adversarial(self):
        rs = np.random.RandomState(17)
        for trial, (N, d, n, T) in enumerate(
                [(3000, 6, 256, 30), (1200, 3, 64, 10), (800, 12, 128, 7)]):
            X = rs.normal(size=(N, d)).astype(np.float32)
            bag = cpu_engine.sample_bags(N, T, n, seed=trial, bootstrap=False)
            fs = cpu_engine.feature_subsets(d, d, T, seed=trial)
            f = cpu_engine.build_forest(X, bag, fs, trial, n, d, d)
            if trial == 1:  # foreign knife-edge splits
                internal = f.feature >= 0
                v32 = f.value.astype(np.float64)
                nxt = np.nextafter(f.value,
                                   np.float32(np.inf)).astype(np.float64)
                f.value64 = np.where(internal, (v32 + nxt) / 2, f.value64)
                X[:200, 0] = f.value[0, 0]
            if trial == 2:
                X[rs.randint(0, N, 40), rs.randint(0, d, 40)] = np.nan
            a = cpu_engine._path_lengths_fast(f, X)
            b = cpu_engine._path_lengths_ref(f, X)
            np.testing.assert_array_equal(
                a.view(np.int32), b.view(np.int32), err_msg=f"trial {trial}")

    def test_env_forces_reference(self, monkeypatch):
        rs = np.random.RandomState(18)
        X = rs.normal(size=(500, 4)).astype(np.float32)
        bag = cpu_engine.sample_bags(500, 4, 64, seed=1, bootstrap=False)
        fs = cpu_engine.feature_subsets(4, 4, 4, seed=1)
        f = cpu_engine.build_forest(X, bag, fs, 1, 64, 4, 4)
        monkeypatch.setenv("IFA_CPU_WALK_REF", "1")
        a = cpu_engine.path_lengths(f, X)
        monkeypatch.delenv("IFA_CPU_WALK_REF")
        b = cpu_engine.path_lengths(f, X)
        np.testing.assert_array_equal(a.view(np.int32), b.view(np.int32))
