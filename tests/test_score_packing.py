"""CPU tests for the v4 scoring pack: order-preserving key transforms and
depth/leaf folding (no GPU needed — pure numpy mirrors of the device logic).
"""

import math

import numpy as np
import pytest

from isolation_forest_amd.core import cpu_engine
from isolation_forest_amd.core.forest import Forest
from isolation_forest_amd.ops.gpu_engine import (
    _bf16_key_tables,
    _bf16_threshold_keys,
    _key32,
    _node_depths,
    _nodes_packed_v4,
)


def _stage_key16(bits16: np.ndarray) -> np.ndarray:
    """numpy mirror of the device key16() staging transform."""
    b = bits16.astype(np.uint32)
    m = b & np.uint32(0x7FFF)
    k = np.where(b & np.uint32(0x8000), np.uint32(0x7FFF) - m,
                 np.uint32(0x8000) + m).astype(np.uint32)
    k[m == 0] = np.uint32(0x8000)
    k[m > np.uint32(0x7F80)] = np.uint32(0xFFFF)
    return k


def _widen16(k: np.ndarray) -> np.ndarray:
    return (k << np.uint32(16)) | np.uint32(0xFFFF)


def _bf16_bits(x32: np.ndarray) -> np.ndarray:
    """f32 -> bf16 bits by truncation-free round-to-nearest-even (torch's
    conversion); here we only need EXACT bf16 values, so build them by
    truncating then viewing back — inputs in tests are already bf16-exact."""
    return (x32.view(np.uint32) >> np.uint32(16)).astype(np.uint16)


class TestKey32:
    def test_order_preserving_random(self):
        rs = np.random.RandomState(0)
        v = np.concatenate([
            rs.normal(scale=1e-20, size=500).astype(np.float32),
            rs.normal(scale=1e20, size=500).astype(np.float32),
            rs.normal(size=1000).astype(np.float32),
            np.array([0.0, -0.0, np.inf, -np.inf, 1e-45, -1e-45],
                     dtype=np.float32),
        ])
        k = _key32(v)
        order_v = np.argsort(v, kind="stable")
        sv, sk = v[order_v], k[order_v]
        assert np.all(np.diff(sk.astype(np.int64)) * (np.diff(sv) > 0) >= 0)
        # strict: v[i] < v[j]  <=>  k[i] < k[j]
        i, j = np.triu_indices(200)
        vi, vj = v[:200][i], v[:200][j]
        ki, kj = k[:200][i], k[:200][j]
        np.testing.assert_array_equal(vi < vj, ki < kj)

    def test_zero_collapse_and_nan(self):
        v = np.array([0.0, -0.0, np.nan, -np.nan], dtype=np.float32)
        k = _key32(v)
        assert k[0] == k[1] == 0x80000000
        assert k[2] == k[3] == 0xFFFFFFFF


class TestBf16Keys:
    def test_threshold_semantics_random(self):
        """(x < s) in f32 must equal (widened key(x) < threshold_key(s))
        for every bf16-exact x and arbitrary f32 split s."""
        rs = np.random.RandomState(1)
        x32 = rs.normal(size=4000).astype(np.float32)
        # make x bf16-exact (truncate mantissa)
        x32 = ((x32.view(np.uint32) & np.uint32(0xFFFF0000))).view(np.float32)
        x32 = np.concatenate([
            x32, np.array([0.0, -0.0, np.inf, -np.inf], dtype=np.float32)])
        s = np.concatenate([
            rs.normal(size=200).astype(np.float32),
            x32[:50],  # splits exactly at data points
            np.nextafter(x32[:50], np.float32(np.inf)),
            np.array([0.0, 1e-30, -1e-30], dtype=np.float32),
        ])
        t = _bf16_threshold_keys(s)
        xk = _widen16(_stage_key16(_bf16_bits(x32)))
        for si, ti in zip(s, t):
            expect = x32 < si
            got = xk < np.uint32(ti)
            np.testing.assert_array_equal(got, expect, err_msg=f"s={si}")

    def test_sentinel_never_left(self):
        vals, keys = _bf16_key_tables()
        t = _bf16_threshold_keys(np.array([np.finfo(np.float32).max],
                                          dtype=np.float32))
        sentinel = np.uint32(0xFFFFFFFF)
        assert not sentinel < t[0]

    def test_tables_monotone(self):
        vals, keys = _bf16_key_tables()
        assert np.all(np.diff(vals) > 0)
        assert np.all(np.diff(keys.astype(np.int64)) > 0)


class TestPackedV4:
    def _forest(self, seed=5, T=6, n=128, d=7):
        rs = np.random.RandomState(seed)
        X = rs.normal(size=(3000, d)).astype(np.float32)
        bag = cpu_engine.sample_bags(3000, T, n, seed=seed, bootstrap=False)
        fs = cpu_engine.feature_subsets(d, d, T, seed=seed)
        return X, cpu_engine.build_forest(X, bag, fs, seed, n, d, d)

    def test_depths_match_recursion(self):
        X, forest = self._forest()
        depth = _node_depths(forest.feature, forest.right)

        def walk(t, i, dep):
            assert depth[t, i] == dep
            if forest.feature[t, i] >= 0:
                walk(t, i + 1, dep + 1)
                walk(t, forest.right[t, i], dep + 1)

        for t in range(forest.num_trees):
            walk(t, 0, 0)

    def test_leaf_values_fold_depth(self):
        X, forest = self._forest()
        packed, ncount, max_depth = _nodes_packed_v4(forest, d_sentinel=7,
                                                     bf16=False)
        assert packed.shape[0] % 4 == 0
        assert ncount.shape[0] == packed.shape[0]
        depth = _node_depths(forest.feature, forest.right)
        leaf = forest.feature == Forest.LEAF
        w1 = packed[: forest.num_trees, :, 1].view(np.uint32)
        got = w1[leaf].view(np.float32)
        expect = (depth[leaf].astype(np.float32)
                  + forest.value[leaf].astype(np.float32))
        np.testing.assert_array_equal(got, expect)
        assert max_depth == int(depth[leaf | (forest.feature >= 0)].max())

    def test_leaf_self_loop_and_sentinel(self):
        X, forest = self._forest()
        packed, _, _ = _nodes_packed_v4(forest, d_sentinel=7, bf16=True)
        leaf = forest.feature == Forest.LEAF
        w0 = packed[: forest.num_trees, :, 0]
        ids = np.broadcast_to(
            np.arange(forest.feature.shape[1], dtype=np.int32)[None, :],
            forest.feature.shape)
        assert np.all((w0[leaf] & 0xFFF) == 7)
        assert np.all(((w0[leaf] >> 12) & 0x7FFF) == ids[leaf])

    def test_walk_simulation_matches_oracle(self):
        """Simulate the v4 kernel walk in numpy (keys, self-loops, fixed
        trips) and compare path sums bitwise against cpu_engine."""
        X, forest = self._forest(seed=11, T=8, n=256, d=5)
        oracle = cpu_engine.path_lengths(forest, X)
        packed, ncount, max_depth = _nodes_packed_v4(forest, d_sentinel=5,
                                                     bf16=False)
        Tn = forest.num_trees
        w0 = packed[:, :, 0]
        w1 = packed[:, :, 1].view(np.uint32)
        keys = _key32(X.reshape(-1)).reshape(X.shape)
        keys = np.concatenate(
            [keys, np.full((X.shape[0], 1), 0xFFFFFFFF, dtype=np.uint32)],
            axis=1)
        total = np.zeros(X.shape[0], dtype=np.float32)
        for t in range(Tn):
            cur = np.zeros(X.shape[0], dtype=np.int64)
            for _ in range(max_depth):
                f = w0[t, cur] & 0xFFF
                right = (w0[t, cur] >> 12) & 0x7FFF
                x = keys[np.arange(X.shape[0]), f]
                cur = np.where(x < w1[t, cur], cur + 1, right)
            total = (total + w1[t, cur].view(np.float32)).astype(np.float32)
        np.testing.assert_array_equal(total.view(np.int32),
                                      oracle.view(np.int32))


class TestThresholdKeyClosedForm:
    def test_matches_table_reference(self):
        from isolation_forest_amd.ops.gpu_engine import (
            _bf16_threshold_keys, _bf16_threshold_keys_table)

        rs = np.random.RandomState(3)
        s = np.concatenate([
            rs.normal(size=5000).astype(np.float32),
            rs.normal(scale=1e-40, size=500).astype(np.float32),
            rs.normal(scale=1e38, size=500).astype(np.float32),
            # exact bf16 values and their neighbours
            ((rs.randint(0, 1 << 16, size=2000).astype(np.uint32) << 16)
             .view(np.float32)),
            np.array([0.0, -0.0, 1e-45, -1e-45,
                      np.finfo(np.float32).max, -np.finfo(np.float32).max,
                      np.inf, -np.inf], dtype=np.float32),
        ])
        s = s[np.isfinite(s) | np.isinf(s)]
        s = s[~np.isnan(s)]
        np.testing.assert_array_equal(
            _bf16_threshold_keys(s), _bf16_threshold_keys_table(s))


class TestWidePacking:
    """Wide int4 node records (score_forest_wide / score_extended_wide):
    same key/leaf-value math as the packed v4 format, fields unpacked."""

    def _forest(self):
        rs = np.random.RandomState(9)
        X = rs.normal(size=(3000, 6)).astype(np.float32)
        bag = cpu_engine.sample_bags(3000, 8, 256, seed=4, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 8, seed=4)
        return cpu_engine.build_forest(X, bag, fs, 4, 256, 6, 6)

    def test_fields_match_v4_semantics(self):
        from isolation_forest_amd.ops.gpu_engine import (
            _nodes_packed_v4, _nodes_packed_wide)

        forest = self._forest()
        for bf16 in (False, True):
            v4, _, h4 = _nodes_packed_v4(forest, 6, bf16)
            wide, hw_ = _nodes_packed_wide(forest, bf16)
            assert hw_ == h4
            T, mn = forest.feature.shape
            internal = forest.feature >= 0
            leaf = forest.feature == -1
            # internal: feat/right fields equal the v4 bit fields
            np.testing.assert_array_equal(
                wide[..., 0][internal], v4[:T][..., 0][internal] & 0xFFF)
            np.testing.assert_array_equal(
                wide[..., 1][internal],
                (v4[:T][..., 0][internal] >> 12) & 0x7FFF)
            # keys/leaf values identical
            np.testing.assert_array_equal(
                wide[..., 2][internal | leaf],
                v4[:T][..., 1][internal | leaf])
            # leaves and pad self-loop
            ids = np.broadcast_to(np.arange(mn, dtype=np.int32), (T, mn))
            np.testing.assert_array_equal(wide[..., 1][~internal],
                                          ids[~internal])

    def test_packed_fits_predicate(self):
        from isolation_forest_amd.ops.gpu_engine import _packed_fits

        forest = self._forest()
        assert _packed_fits(forest, 6)
        assert _packed_fits(forest, 4094)
        assert not _packed_fits(forest, 4095)

    def test_eif_wide_fields(self):
        from isolation_forest_amd.ops.gpu_engine import _eif_packed_wide

        rs = np.random.RandomState(10)
        X = rs.normal(size=(2000, 5)).astype(np.float32)
        bag = cpu_engine.sample_bags(2000, 4, 128, seed=5, bootstrap=False)
        fs = cpu_engine.feature_subsets(5, 5, 4, seed=5)
        forest = cpu_engine.build_extended_forest(X, bag, fs, 5, 128, 5, 5, 4)
        wide = _eif_packed_wide(forest)
        internal = forest.feature >= 0
        assert (wide[..., 0][internal] == 0).all()
        assert (wide[..., 0][~internal] == -1).all()
        np.testing.assert_array_equal(
            wide[..., 1][internal], forest.right[internal])
        np.testing.assert_array_equal(
            wide[..., 2].view(np.float32)[internal],
            forest.value.astype(np.float32)[internal])


class TestEif0MirrorPack:
    """extensionLevel-0 pack: exact key thresholds + mirrored subtrees
    must make the UNMODIFIED v4 walk reproduce the oracle bitwise. The
    v4 walk is simulated in numpy here (same simulation proven against
    the GPU kernel in TestPackedV4); GPU execution is covered by
    tests/test_gpu.py::TestEif0Route."""

    def _simulate_and_compare(self, forest, X, bf16):
        import torch as _t

        from isolation_forest_amd.ops.gpu_engine import (
            _bf16_key_tables, _eif0_packed_v4, _key32)

        if bf16:
            X = _t.from_numpy(X).to(_t.bfloat16).float().numpy()
        oracle = cpu_engine.path_lengths_extended(forest, X)
        d = X.shape[1]
        packed, ncount, max_depth = _eif0_packed_v4(forest, d, bf16)
        w0 = packed[:, :, 0]
        w1 = packed[:, :, 1].view(np.uint32)
        if bf16:
            vals, keys = _bf16_key_tables()
            idx = np.searchsorted(vals, X.reshape(-1))
            k = keys[np.minimum(idx, len(keys) - 1)].astype(np.uint32)
            keys_m = ((k.astype(np.uint64) << 16) | 0xFFFF).astype(
                np.uint64).reshape(X.shape)
        else:
            keys_m = _key32(X.reshape(-1)).astype(np.uint64).reshape(X.shape)
        keys_m = np.concatenate(
            [keys_m, np.full((X.shape[0], 1), 0xFFFFFFFF,
                             dtype=np.uint64)], axis=1)
        total = np.zeros(X.shape[0], dtype=np.float32)
        for t in range(forest.num_trees):
            cur = np.zeros(X.shape[0], dtype=np.int64)
            for _ in range(max_depth):
                f = w0[t, cur] & 0xFFF
                right = (w0[t, cur] >> 12) & 0x7FFF
                x = keys_m[np.arange(X.shape[0]), f]
                cur = np.where(x < w1[t, cur].astype(np.uint64),
                               cur + 1, right)
            total = (total + w1[t, cur].view(np.float32)).astype(np.float32)
        np.testing.assert_array_equal(total.view(np.int32),
                                      oracle.view(np.int32))

    def _ext0_forest(self, seed, foreign=False, knife=False):
        rs = np.random.RandomState(seed)
        X = rs.normal(size=(2000, 5)).astype(np.float32)
        bag = cpu_engine.sample_bags(2000, 8, 128, seed=seed,
                                     bootstrap=False)
        fs = cpu_engine.feature_subsets(5, 5, 8, seed=seed)
        forest = cpu_engine.build_extended_forest(X, bag, fs, seed, 128, 5,
                                                  5, 0)
        internal = forest.feature >= 0
        if foreign:
            w = rs.choice([0.5, -3.7, 1e-3, -1e4, 2.0, -1e-20],
                          internal.sum()).astype(np.float32)
            forest.hyper_w[..., 0][internal] = w
        if knife:
            ik = np.argwhere(internal)
            pick = ik[rs.randint(0, len(ik), size=300)]
            with np.errstate(all="ignore"):
                for r, (t, n) in enumerate(pick):
                    X[r, forest.hyper_idx[t, n, 0]] = np.float32(
                        forest.value[t, n] / forest.hyper_w[t, n, 0])
        return forest, X

    def test_native_weights_f32(self):
        forest, X = self._ext0_forest(1)
        self._simulate_and_compare(forest, X, bf16=False)

    def test_native_weights_bf16(self):
        forest, X = self._ext0_forest(2)
        self._simulate_and_compare(forest, X, bf16=True)

    def test_foreign_weights_knife_edge_f32(self):
        forest, X = self._ext0_forest(3, foreign=True, knife=True)
        self._simulate_and_compare(forest, X, bf16=False)

    def test_foreign_weights_bf16(self):
        forest, X = self._ext0_forest(4, foreign=True)
        self._simulate_and_compare(forest, X, bf16=True)

    def test_specials(self):
        """zeros, denormals and infs in the data route exactly."""
        forest, X = self._ext0_forest(5, foreign=True)
        rs = np.random.RandomState(6)
        sp = np.array([0.0, -0.0, np.inf, -np.inf, 1e-45, -1e-45],
                      dtype=np.float32)
        X[rs.randint(0, len(X), 200), rs.randint(0, 5, 200)] = rs.choice(
            sp, 200)
        self._simulate_and_compare(forest, X, bf16=False)

    def test_mirror_preserves_depths(self):
        from isolation_forest_amd.ops.gpu_engine import (
            _eif0_packed_v4, _node_depths)

        forest, _ = self._ext0_forest(7, foreign=True)
        _, _, max_depth = _eif0_packed_v4(forest, 5, False)
        depth = _node_depths(forest.feature, forest.right)
        assert max_depth == int(depth[(forest.feature >= 0)
                                      | (forest.feature == -1)].max())
