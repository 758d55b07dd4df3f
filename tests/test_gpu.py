"""GPU tests (MI355X): bitwise CPU<->GPU parity, bf16 path, end-to-end quality.

All marked @pytest.mark.gpu; they run on the GPU box via
`python -m pytest tests -m gpu` and are skipped where no GPU exists.
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from isolation_forest_amd import ExtendedIsolationForest, IsolationForest
from isolation_forest_amd.core import cpu_engine
from tests.conftest import auroc


@pytest.fixture(scope="module")
def dev():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from isolation_forest_amd.ops import load_extension

    load_extension()  # hard error if missing on a GPU box
    return torch.device("cuda:0")


def make_data(n=5000, d=8, seed=3):
    rs = np.random.RandomState(seed)
    return rs.normal(size=(n, d)).astype(np.float32)


class TestBagGather:
    def test_matches_cpu_gather(self, dev):
        from isolation_forest_amd.ops import load_extension

        ext = load_extension()
        X = make_data(1000, 8)
        idx = cpu_engine.sample_bags(1000, 10, 64, seed=1, bootstrap=False)
        Xt = torch.from_numpy(X).to(dev)
        bags = ext.bag_gather(Xt, torch.from_numpy(idx).to(dev))
        expect = X[idx]  # [10, 64, 8]
        np.testing.assert_array_equal(bags.cpu().numpy(), expect)

    def test_bf16_gather(self, dev):
        from isolation_forest_amd.ops import load_extension

        ext = load_extension()
        X = make_data(500, 4)
        Xb = torch.from_numpy(X).to(dev).to(torch.bfloat16)
        idx = cpu_engine.sample_bags(500, 5, 32, seed=2, bootstrap=False)
        bags = ext.bag_gather(Xb, torch.from_numpy(idx).to(dev))
        expect = Xb.float().cpu().numpy()[idx]
        np.testing.assert_array_equal(bags.cpu().numpy(), expect)


class TestStructuralParity:
    """The strongest GPU check: GPU-built forests are BIT-IDENTICAL to the
    CPU oracle's for the same seed."""

    @pytest.mark.parametrize("n,d,k,T", [(256, 8, 8, 16), (128, 16, 5, 8), (1024, 4, 4, 4)])
    def test_standard_build_identical(self, dev, n, d, k, T):
        from isolation_forest_amd.ops import gpu_engine
        from isolation_forest_amd.utils.params import ResolvedParams

        X = make_data(8000, d)
        bag = cpu_engine.sample_bags(8000, T, n, seed=9, bootstrap=False)
        fs = cpu_engine.feature_subsets(d, k, T, seed=9)
        cpu_f = cpu_engine.build_forest(X, bag, fs, 9, n, k, d)
        rp = ResolvedParams(num_samples=n, num_features=k, total_rows=8000,
                            total_features=d)
        gpu_f = gpu_engine.build_forest(torch.from_numpy(X).to(dev), bag, fs, 9, rp)
        np.testing.assert_array_equal(gpu_f.node_count, cpu_f.node_count)
        np.testing.assert_array_equal(gpu_f.feature, cpu_f.feature)
        np.testing.assert_array_equal(gpu_f.right, cpu_f.right)
        np.testing.assert_array_equal(gpu_f.num_instances, cpu_f.num_instances)
        np.testing.assert_array_equal(
            gpu_f.value.view(np.int32), cpu_f.value.view(np.int32)
        )

    @pytest.mark.parametrize("ext_level", [0, 2, 7])
    def test_extended_build_identical(self, dev, ext_level):
        from isolation_forest_amd.ops import gpu_engine
        from isolation_forest_amd.utils.params import ResolvedParams

        n, d, T = 256, 8, 8
        X = make_data(6000, d)
        bag = cpu_engine.sample_bags(6000, T, n, seed=4, bootstrap=False)
        fs = cpu_engine.feature_subsets(d, d, T, seed=4)
        cpu_f = cpu_engine.build_extended_forest(X, bag, fs, 4, n, d, d, ext_level)
        rp = ResolvedParams(num_samples=n, num_features=d, total_rows=6000,
                            total_features=d, extension_level=ext_level)
        gpu_f = gpu_engine.build_extended_forest(
            torch.from_numpy(X).to(dev), bag, fs, 4, rp
        )
        np.testing.assert_array_equal(gpu_f.node_count, cpu_f.node_count)
        np.testing.assert_array_equal(gpu_f.feature, cpu_f.feature)
        np.testing.assert_array_equal(gpu_f.right, cpu_f.right)
        np.testing.assert_array_equal(gpu_f.num_instances, cpu_f.num_instances)
        np.testing.assert_array_equal(gpu_f.hyper_idx, cpu_f.hyper_idx)
        np.testing.assert_array_equal(
            gpu_f.hyper_w.view(np.int32), cpu_f.hyper_w.view(np.int32)
        )
        np.testing.assert_array_equal(
            gpu_f.value.view(np.int32), cpu_f.value.view(np.int32)
        )
        # offsets are doubles computed in identical order
        np.testing.assert_array_equal(gpu_f.offset64, cpu_f.offset64)


class TestScoringParity:
    def test_path_sums_bitwise(self, dev):
        from isolation_forest_amd.ops import gpu_engine

        X = make_data(4000, 8)
        bag = cpu_engine.sample_bags(4000, 32, 256, seed=2, bootstrap=False)
        fs = cpu_engine.feature_subsets(8, 8, 32, seed=2)
        forest = cpu_engine.build_forest(X, bag, fs, 2, 256, 8, 8)
        cpu_ps = cpu_engine.path_lengths(forest, X)

        model = IsolationForest(numEstimators=32).fit(X[:600])  # shell for cache
        model.forest = forest
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_forest(
            model, torch.from_numpy(X).to(dev), finalize=False
        )
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32)
        )

    def test_scores_close(self, dev):
        X = make_data(3000, 8)
        model = IsolationForest(numEstimators=50, randomSeed=6).fit(X)
        cpu_scores = model.score(torch.from_numpy(X)).numpy()
        gpu_scores = model.score(torch.from_numpy(X).to(dev)).cpu().numpy()
        np.testing.assert_allclose(gpu_scores, cpu_scores, rtol=0, atol=2e-7)

    def test_extended_dense_path_close(self, dev):
        """nnz == d engages the dense fast path, whose dot reassociates the
        accumulation — tolerance, not bitwise (build stays bitwise)."""
        from isolation_forest_amd.ops import gpu_engine

        X = make_data(3000, 8)
        bag = cpu_engine.sample_bags(3000, 16, 128, seed=5, bootstrap=False)
        fs = cpu_engine.feature_subsets(8, 8, 16, seed=5)
        forest = cpu_engine.build_extended_forest(X, bag, fs, 5, 128, 8, 8, 7)
        cpu_ps = cpu_engine.path_lengths_extended(forest, X)
        model = ExtendedIsolationForest(numEstimators=16).fit(X[:600])
        model.forest = forest
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_extended_forest(
            model, torch.from_numpy(X).to(dev), finalize=False
        )
        diff = np.abs(gpu_ps.cpu().numpy() - cpu_ps)
        # reassociated dots may flip knife-edge path decisions for isolated
        # rows; everything else must agree to float noise
        assert (diff > 1e-3).sum() <= 3
        assert np.quantile(diff, 0.999) < 1e-3

    def test_extended_dense_v3_bf16_path_close(self, dev):
        """bf16 rows route to dense v3 (packed-bf16 weights + v_dot2c).
        Oracle: strict-order CPU walk with the SAME RNE-bf16-rounded
        weights — remaining diffs are reassociation-level flips only."""
        import copy

        from isolation_forest_amd.ops import gpu_engine
        from isolation_forest_amd.utils.det_math import bf16_round

        rs = np.random.RandomState(11)
        Xb = torch.from_numpy(
            rs.normal(size=(3000, 8)).astype(np.float32)
        ).to(torch.bfloat16)
        X = Xb.float().numpy()  # exact f32 of the bf16 rows
        bag = cpu_engine.sample_bags(3000, 16, 128, seed=5, bootstrap=False)
        fs = cpu_engine.feature_subsets(8, 8, 16, seed=5)
        forest = cpu_engine.build_extended_forest(X, bag, fs, 5, 128, 8, 8, 7)
        f_b = copy.copy(forest)
        f_b.hyper_w = bf16_round(forest.hyper_w)
        cpu_ps = cpu_engine.path_lengths_extended(f_b, X)
        model = ExtendedIsolationForest(numEstimators=16).fit(X[:600])
        model.forest = forest
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_extended_forest(
            model, Xb.to(dev), finalize=False
        )
        diff = np.abs(gpu_ps.cpu().numpy() - cpu_ps)
        assert (diff > 1e-3).sum() <= 5
        assert np.quantile(diff, 0.99) < 1e-3

    def test_extended_dense_v3_matches_v2_statistically(self, dev):
        """v3 (bf16 weights) vs v2 (f32 weights) on the same bf16 rows:
        path sums agree except for knife-edge walk flips; scores stay
        rank-consistent (the documented bf16-weight precision cost)."""
        import os as _os

        from isolation_forest_amd.ops import gpu_engine

        rs = np.random.RandomState(12)
        Xb = torch.from_numpy(
            rs.normal(size=(20000, 32)).astype(np.float32)
        ).to(torch.bfloat16)
        X32 = Xb.float().numpy()
        model = ExtendedIsolationForest(numEstimators=64, randomSeed=3).fit(
            X32
        )
        model._gpu_forest_cache = {}
        s3 = gpu_engine.score_extended_forest(
            model, Xb.to(dev), finalize=True
        ).cpu().numpy()
        _os.environ["IFA_EIF_DENSE_V2"] = "1"
        try:
            model._gpu_forest_cache = {}
            s2 = gpu_engine.score_extended_forest(
                model, Xb.to(dev), finalize=True
            ).cpu().numpy()
        finally:
            del _os.environ["IFA_EIF_DENSE_V2"]
        # scores: small perturbation, high rank agreement
        assert np.abs(s3 - s2).max() < 0.05
        assert np.abs(s3 - s2).mean() < 0.005
        r2 = np.argsort(np.argsort(s2))
        r3 = np.argsort(np.argsort(s3))
        corr = np.corrcoef(r2, r3)[0, 1]
        assert corr > 0.999

    def test_extended_path_sums_bitwise(self, dev):
        from isolation_forest_amd.ops import gpu_engine

        X = make_data(2000, 6)
        bag = cpu_engine.sample_bags(2000, 16, 128, seed=3, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 16, seed=3)
        forest = cpu_engine.build_extended_forest(X, bag, fs, 3, 128, 6, 6, 5)
        cpu_ps = cpu_engine.path_lengths_extended(forest, X)

        model = ExtendedIsolationForest(numEstimators=16).fit(X[:600])
        model.forest = forest
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_extended_forest(
            model, torch.from_numpy(X).to(dev), finalize=False
        )
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32)
        )


class TestEndToEndGPU:
    def test_fit_transform_on_device(self, dev):
        rs = np.random.RandomState(7)
        d = 8
        inl = rs.normal(size=(20000, d)).astype(np.float32)
        out = rs.uniform(-10, 12, size=(400, d)).astype(np.float32)
        X = np.concatenate([inl, out])
        y = np.concatenate([np.zeros(20000), np.ones(400)])
        Xt = torch.from_numpy(X).to(dev)
        model = IsolationForest(
            numEstimators=100, contamination=0.02, contaminationError=0.0
        ).fit(Xt)
        res = model.transform(Xt)
        scores = res["outlierScore"].cpu().numpy()
        assert auroc(y, scores) > 0.9
        frac = float(res["predictedLabel"].cpu().float().mean())
        assert frac == pytest.approx(0.02, abs=0.005)

    def test_fit_transform_bf16(self, dev):
        rs = np.random.RandomState(8)
        X = rs.normal(size=(30000, 32)).astype(np.float32)
        Xb = torch.from_numpy(X).to(dev).to(torch.bfloat16)
        model = IsolationForest(numEstimators=64).fit(Xb)
        scores = model.transform(Xb)["outlierScore"]
        assert scores.is_cuda
        assert float(scores.min()) > 0 and float(scores.max()) < 1
        # bf16 GPU fit == CPU fit on the bf16-rounded matrix (same seed)
        Xr = Xb.float().cpu().numpy()
        cpu_model = IsolationForest(numEstimators=64).fit(Xr)
        for t in range(0, 64, 16):
            assert (
                model.forest.tree_to_string(t) == cpu_model.forest.tree_to_string(t)
            )

    def test_extended_fit_on_device(self, dev):
        rs = np.random.RandomState(9)
        X = torch.from_numpy(rs.normal(size=(10000, 6)).astype(np.float32)).to(dev)
        model = ExtendedIsolationForest(numEstimators=50).fit(X)
        scores = model.transform(X)["outlierScore"]
        assert scores.is_cuda and scores.shape[0] == 10000

    def test_gpu_model_roundtrip_to_cpu(self, dev, tmp_path):
        X = make_data(3000, 8)
        Xt = torch.from_numpy(X).to(dev)
        model = IsolationForest(numEstimators=20, contamination=0.05).fit(Xt)
        path = str(tmp_path / "gpu_model")
        model.save(path)
        from isolation_forest_amd import IsolationForestModel

        loaded = IsolationForestModel.load(path)
        s_gpu = model.score(Xt).cpu().numpy()
        s_cpu = loaded.score(torch.from_numpy(X)).numpy()
        np.testing.assert_allclose(s_cpu, s_gpu, rtol=0, atol=2e-7)


class TestRealDataQualityGPU:
    """Real-data AUROC gates on the GPU path (VERDICT r01 Missing #4):
    in-repo ODDS fixtures make these run on the driver's GPU box, where
    synthetic-only gates (bench AUROC 1.0) are non-discriminating.
    Reference bars: IsolationForestTest.scala:77-88 (mammography 0.86),
    :229-236 (shuttle > 0.99)."""

    def test_mammography_auroc_gpu(self, dev, mammography):
        X, y = mammography
        Xt = torch.from_numpy(X).to(dev)
        model = IsolationForest(
            numEstimators=100, maxSamples=256.0, randomSeed=1
        ).fit(Xt)
        scores = model.score(Xt).cpu().numpy()
        assert auroc(y, scores) == pytest.approx(0.86, abs=0.02)

    def test_mammography_auroc_gpu_extended(self, dev, mammography):
        X, y = mammography
        Xt = torch.from_numpy(X).to(dev)
        model = ExtendedIsolationForest(
            numEstimators=100, maxSamples=256.0, randomSeed=1
        ).fit(Xt)
        scores = model.score(Xt).cpu().numpy()
        assert auroc(y, scores) == pytest.approx(0.86, abs=0.02)

    def test_shuttle_auroc_gpu(self, dev, shuttle):
        X, y = shuttle
        Xt = torch.from_numpy(X).to(dev)
        model = IsolationForest(
            numEstimators=100, maxSamples=256.0, randomSeed=1
        ).fit(Xt)
        scores = model.score(Xt).cpu().numpy()
        assert auroc(y, scores) > 0.99
        assert float(scores[y == 1].mean()) == pytest.approx(0.61, abs=0.02)
        assert float(scores[y == 0].mean()) == pytest.approx(0.41, abs=0.02)


class TestDevicePacking:
    """The torch-side scoring pack (built from raw device build outputs)
    must be BITWISE identical to the host numpy pack."""

    def test_standard_pack_matches_host(self, dev):
        from isolation_forest_amd.ops import gpu_engine

        X = torch.from_numpy(make_data(20000, 12, seed=21)).to(dev)
        model = IsolationForest(numEstimators=64, randomSeed=5).fit(X)
        forest = model.forest
        raw = forest._device_raw
        for bf16 in (True, False):
            aos_dev, nc_dev, h_dev = gpu_engine._nodes_packed_v4_device(
                raw, forest.num_trees, 12, bf16)
            aos_host, nc_host, h_host = gpu_engine._nodes_packed_v4(
                forest, 12, bf16)
            np.testing.assert_array_equal(aos_dev.cpu().numpy(), aos_host)
            np.testing.assert_array_equal(nc_dev.cpu().numpy(), nc_host)
            assert h_dev == h_host

    @pytest.mark.parametrize("ext_level", [2, 5])
    def test_extended_pack_matches_host(self, dev, ext_level):
        from isolation_forest_amd.ops import gpu_engine

        X = torch.from_numpy(make_data(15000, 6, seed=22)).to(dev)
        model = ExtendedIsolationForest(
            numEstimators=32, randomSeed=7, extensionLevel=ext_level).fit(X)
        forest = model.forest
        raw = forest._device_raw
        aos_dev, vals_dev, hw_dev, h_dev = gpu_engine._eif_dense_packed_device(
            raw, 8)
        aos_host, vals_host, hw_host, h_host = gpu_engine._eif_dense_packed(
            forest, 8)
        np.testing.assert_array_equal(aos_dev.cpu().numpy(), aos_host)
        np.testing.assert_array_equal(vals_dev.cpu().numpy(), vals_host)
        np.testing.assert_array_equal(hw_dev.cpu().numpy(), hw_host)
        assert h_dev == h_host

    def test_sparse_densified_route_matches_oracle(self, dev):
        """nnz >= 6 but < d routes through the densified dense kernel;
        scores must match the CPU oracle to walk-decision tolerance."""
        X = make_data(20000, 12, seed=24)
        model = ExtendedIsolationForest(
            numEstimators=40, randomSeed=3, extensionLevel=7).fit(
            torch.from_numpy(X))
        cpu_scores = model.score(torch.from_numpy(X)).numpy()
        gpu_scores = model.score(torch.from_numpy(X).to(dev)).cpu().numpy()
        diff = np.abs(gpu_scores - cpu_scores)
        assert (diff > 1e-3).sum() <= 3
        assert np.quantile(diff, 0.999) < 1e-3

    def test_scores_identical_both_paths(self, dev):
        X = torch.from_numpy(make_data(30000, 10, seed=23)).to(dev)
        model = IsolationForest(numEstimators=64, randomSeed=9).fit(X)
        s_dev_pack = model.score(X).cpu().numpy()
        # force the host packing path (drop raw AND the rebuildable depths)
        del model.forest._device_raw
        model.forest.depth_np = None
        model._gpu_forest_cache = {}
        s_host_pack = model.score(X).cpu().numpy()
        np.testing.assert_array_equal(
            s_dev_pack.view(np.int32), s_host_pack.view(np.int32))


class TestSparseV2:
    def test_sparse_path_sums_bitwise(self, dev):
        """nnz <= 5 routes through the fixed-trip sparse v2 kernel, which
        keeps the oracle's strict j-order dot: bitwise path sums."""
        from isolation_forest_amd.ops import gpu_engine

        X = make_data(3000, 6, seed=31)
        bag = cpu_engine.sample_bags(3000, 24, 128, seed=8, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 24, seed=8)
        forest = cpu_engine.build_extended_forest(X, bag, fs, 8, 128, 6, 6, 3)
        cpu_ps = cpu_engine.path_lengths_extended(forest, X)
        model = ExtendedIsolationForest(numEstimators=24).fit(X[:600])
        model.forest = forest
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_extended_forest(
            model, torch.from_numpy(X).to(dev), finalize=False
        )
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32)
        )

    def test_sparse_end_to_end_gpu_fit(self, dev):
        X = torch.from_numpy(make_data(20000, 10, seed=32)).to(dev)
        model = ExtendedIsolationForest(
            numEstimators=50, extensionLevel=2, randomSeed=4).fit(X)
        scores = model.score(X)
        cpu_scores = model.score(X.float().cpu()).numpy()
        np.testing.assert_array_equal(
            scores.cpu().numpy().view(np.int32), cpu_scores.view(np.int32))


class TestWideFeatureFallbacks:
    """d large enough that row tiles exceed LDS exercises the global-X
    walk (key transform on the fly) and the general EIF kernel."""

    def test_standard_wide_d_bitwise(self, dev):
        from isolation_forest_amd.ops import gpu_engine

        d = 700  # bf16 row tile > 150 KB => ROWS_LDS=false path
        X = make_data(3000, d, seed=41)
        bag = cpu_engine.sample_bags(3000, 8, 128, seed=6, bootstrap=False)
        fs = cpu_engine.feature_subsets(d, 32, 8, seed=6)
        forest = cpu_engine.build_forest(X, bag, fs, 6, 128, 32, d)
        cpu_ps = cpu_engine.path_lengths(forest, X)
        model = IsolationForest(numEstimators=8).fit(X[:500])
        model.forest = forest
        model._gpu_forest_cache = {}
        for dt in (torch.float32, torch.bfloat16):
            Xt = torch.from_numpy(X).to(dev).to(dt)
            ref = cpu_ps if dt == torch.float32 else cpu_engine.path_lengths(
                forest, Xt.float().cpu().numpy())
            gpu_ps = gpu_engine.score_forest(model, Xt, finalize=False)
            np.testing.assert_array_equal(
                gpu_ps.cpu().numpy().view(np.int32), ref.view(np.int32))

    def test_extended_wide_d_small_nnz(self, dev):
        X = torch.from_numpy(make_data(20000, 40, seed=42)).to(dev)
        model = ExtendedIsolationForest(
            numEstimators=20, extensionLevel=2, randomSeed=5).fit(X)
        s_gpu = model.score(X).cpu().numpy()
        s_cpu = model.score(X.float().cpu()).numpy()
        np.testing.assert_array_equal(
            s_gpu.view(np.int32), s_cpu.view(np.int32))

    def test_extended_wide_d_wide_nnz(self, dev):
        """d=40 > 32 with nnz=40: round 2 routes this to the D=64 dense
        band (f32 rows -> v2) — reassociation-tolerance contract."""
        X = torch.from_numpy(make_data(15000, 40, seed=43)).to(dev)
        model = ExtendedIsolationForest(numEstimators=12, randomSeed=6).fit(X)
        s_gpu = model.score(X).cpu().numpy()
        s_cpu = model.score(X.float().cpu()).numpy()
        diff = np.abs(s_gpu - s_cpu)
        assert np.quantile(diff, 0.999) < 1e-3


class TestWideDenseBands:
    """Round-2 dense bands D=64/128: fully-extended EIF beyond d=32 no
    longer falls to the general kernel (which measured 8.2M rows/s at
    d=64 — profiles/r02_eif_v3.md)."""

    def _parity(self, dev, d, dtype):
        import copy

        from isolation_forest_amd.ops import gpu_engine
        from isolation_forest_amd.utils.det_math import bf16_round

        rs = np.random.RandomState(50 + d)
        X32 = rs.normal(size=(3000, d)).astype(np.float32)
        if dtype == torch.bfloat16:
            Xt = torch.from_numpy(X32).to(torch.bfloat16)
            X = Xt.float().numpy()
        else:
            Xt = torch.from_numpy(X32)
            X = X32
        bag = cpu_engine.sample_bags(3000, 8, 128, seed=d, bootstrap=False)
        fs = cpu_engine.feature_subsets(d, d, 8, seed=d)
        forest = cpu_engine.build_extended_forest(X, bag, fs, d, 128, d, d,
                                                  d - 1)
        if dtype == torch.bfloat16:
            f_b = copy.copy(forest)
            f_b.hyper_w = bf16_round(forest.hyper_w)
            cpu_ps = cpu_engine.path_lengths_extended(f_b, X)
        else:
            cpu_ps = cpu_engine.path_lengths_extended(forest, X)
        model = ExtendedIsolationForest(numEstimators=8).fit(X[:500])
        model.forest = forest
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_extended_forest(model, Xt.to(dev),
                                                  finalize=False)
        diff = np.abs(gpu_ps.cpu().numpy() - cpu_ps)
        assert (diff > 1e-3).sum() <= 5
        assert np.quantile(diff, 0.99) < 1e-3

    def test_d64_bf16_v3(self, dev):
        self._parity(dev, 64, torch.bfloat16)

    def test_d64_f32_v2(self, dev):
        self._parity(dev, 64, torch.float32)

    def test_d128_bf16_v3(self, dev):
        self._parity(dev, 128, torch.bfloat16)

    def test_d128_f32_general_stays_bitwise(self, dev):
        """f32 rows at d=128 exceed v2's LDS: the general strict-order
        kernel keeps the bitwise contract."""
        from isolation_forest_amd.ops import gpu_engine

        rs = np.random.RandomState(99)
        X = rs.normal(size=(2000, 128)).astype(np.float32)
        bag = cpu_engine.sample_bags(2000, 4, 128, seed=9, bootstrap=False)
        fs = cpu_engine.feature_subsets(128, 128, 4, seed=9)
        forest = cpu_engine.build_extended_forest(X, bag, fs, 9, 128, 128,
                                                  128, 127)
        cpu_ps = cpu_engine.path_lengths_extended(forest, X)
        model = ExtendedIsolationForest(numEstimators=4).fit(X[:500])
        model.forest = forest
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_extended_forest(
            model, torch.from_numpy(X).to(dev), finalize=False)
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32))


class TestNativeCLI:
    """tools/native/ifa_score: the torch-free standalone scorer must
    reproduce the Python engine's scores (same walk decisions; leaf
    constants may differ by a libm-vs-numpy float32-log ulp)."""

    def test_cli_matches_engine(self, dev, tmp_path):
        import subprocess
        import os

        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        cli = os.path.join(repo, "tools", "native", "ifa_score")
        if not os.path.exists(cli):
            subprocess.check_call(
                ["bash", os.path.join(repo, "tools", "native", "build.sh")])
        X = make_data(50000, 10, seed=51)
        model = IsolationForest(
            numEstimators=100, contamination=0.02, contaminationError=0.01,
            randomSeed=12).fit(torch.from_numpy(X).to(dev))
        mdir = str(tmp_path / "m")
        model.save(mdir)
        xbin = str(tmp_path / "x.bin")
        X.astype("<f4").tofile(xbin)
        sbin = str(tmp_path / "s.bin")
        lbin = str(tmp_path / "l.bin")
        out = subprocess.run(
            [cli, mdir, xbin, "50000", "10", sbin, "--labels", lbin],
            capture_output=True, text=True, timeout=300)
        assert out.returncode == 0, out.stderr
        cli_scores = np.fromfile(sbin, dtype="<f4")
        eng_scores = model.score(torch.from_numpy(X).to(dev)).cpu().numpy()
        assert np.abs(cli_scores - eng_scores).max() < 1e-6
        labels = np.fromfile(lbin, dtype=np.uint8)
        expect = model.transform(torch.from_numpy(X).to(dev))[
            "predictedLabel"].cpu().numpy()
        np.testing.assert_array_equal(labels, expect.astype(np.uint8))

    def test_cli_bf16_and_snappy(self, dev, tmp_path):
        import subprocess
        import os

        from isolation_forest_amd.persist import model_io

        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        cli = os.path.join(repo, "tools", "native", "ifa_score")
        X = make_data(30000, 8, seed=52)
        Xb = torch.from_numpy(X).to(dev).to(torch.bfloat16)
        model = IsolationForest(numEstimators=50, randomSeed=13).fit(Xb)
        mdir = str(tmp_path / "m")
        model_io.save_model(model, mdir, codec="snappy")
        xbin = str(tmp_path / "x.bin")
        Xb.view(torch.uint16).cpu().numpy().astype("<u2").tofile(xbin)
        sbin = str(tmp_path / "s.bin")
        out = subprocess.run(
            [cli, mdir, xbin, "30000", "8", sbin, "--bf16"],
            capture_output=True, text=True, timeout=300)
        assert out.returncode == 0, out.stderr
        cli_scores = np.fromfile(sbin, dtype="<f4")
        eng_scores = model.score(Xb).cpu().numpy()
        assert np.abs(cli_scores - eng_scores).max() < 1e-6

    def test_cli_extended_models(self, dev, tmp_path):
        import subprocess
        import os

        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        cli = os.path.join(repo, "tools", "native", "ifa_score")
        X = make_data(30000, 12, seed=53)
        for ext_level, tag in [(2, "sparse"), (11, "dense")]:
            model = ExtendedIsolationForest(
                numEstimators=40, extensionLevel=ext_level,
                randomSeed=14).fit(torch.from_numpy(X).to(dev))
            mdir = str(tmp_path / f"eif_{tag}")
            model.save(mdir)
            xbin = str(tmp_path / f"x_{tag}.bin")
            X.astype("<f4").tofile(xbin)
            sbin = str(tmp_path / f"s_{tag}.bin")
            out = subprocess.run([cli, mdir, xbin, "30000", "12", sbin],
                                 capture_output=True, text=True, timeout=300)
            assert out.returncode == 0, (tag, out.stderr)
            cli_scores = np.fromfile(sbin, dtype="<f4")
            eng = model.score(torch.from_numpy(X).to(dev)).cpu().numpy()
            assert np.abs(cli_scores - eng).max() < 1e-6, tag


class TestDeepForests:
    """Large maxSamples => trees too deep for LDS node staging: the
    nodes-from-global kernel variants must stay bitwise."""

    def test_standard_max_samples_4096(self, dev):
        from isolation_forest_amd.ops import gpu_engine

        X = make_data(9000, 6, seed=61)
        bag = cpu_engine.sample_bags(9000, 4, 4096, seed=15, bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 4, seed=15)
        forest = cpu_engine.build_forest(X, bag, fs, 15, 4096, 6, 6)
        cpu_ps = cpu_engine.path_lengths(forest, X)
        model = IsolationForest(numEstimators=4).fit(X[:500])
        model.forest = forest
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_forest(
            model, torch.from_numpy(X).to(dev), finalize=False)
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32))

    def test_extended_max_samples_2048_full_ext(self, dev):
        X = torch.from_numpy(make_data(12000, 6, seed=62)).to(dev)
        model = ExtendedIsolationForest(
            numEstimators=8, maxSamples=2048.0, randomSeed=16).fit(X)
        s_gpu = model.score(X).cpu().numpy()
        s_cpu = model.score(X.float().cpu()).numpy()
        # deep-forest fallback is the strict-order general kernel: bitwise
        np.testing.assert_array_equal(
            s_gpu.view(np.int32), s_cpu.view(np.int32))

    def test_standard_end_to_end_max_samples_8192(self, dev):
        X = torch.from_numpy(make_data(30000, 10, seed=63)).to(dev)
        model = IsolationForest(
            numEstimators=16, maxSamples=8192.0, randomSeed=17).fit(X)
        s_gpu = model.score(X).cpu().numpy()
        s_cpu = model.score(X.float().cpu()).numpy()
        np.testing.assert_array_equal(
            s_gpu.view(np.int32), s_cpu.view(np.int32))


class TestEif0Route:
    """extensionLevel-0 EIF routes through the v4 walk with exact key
    thresholds (no per-visit multiply): bitwise vs the oracle, and
    identical to the sparse v2 route it replaces."""

    def _forest(self, seed=41, foreign=False):
        rs = np.random.RandomState(seed)
        X = rs.normal(size=(4000, 6)).astype(np.float32)
        bag = cpu_engine.sample_bags(4000, 12, 256, seed=seed,
                                     bootstrap=False)
        fs = cpu_engine.feature_subsets(6, 6, 12, seed=seed)
        forest = cpu_engine.build_extended_forest(X, bag, fs, seed, 256, 6,
                                                  6, 0)
        if foreign:
            # foreign-style single-coordinate hyperplanes: arbitrary
            # magnitudes/signs instead of our +-1-normalized weights
            internal = forest.feature >= 0
            w = rs.choice([0.5, -3.7, 1e-3, -1e4, 2.0],
                          internal.sum()).astype(np.float32)
            forest.hyper_w[..., 0][internal] = w
            # plant knife-edge rows exactly at f32(o / w)
            ik = np.argwhere(internal)
            pick = ik[rs.randint(0, len(ik), size=400)]
            for r, (t, n) in enumerate(pick):
                with np.errstate(all="ignore"):
                    X[r, forest.hyper_idx[t, n, 0]] = np.float32(
                        forest.value[t, n] / forest.hyper_w[t, n, 0])
        return forest, X

    def _score(self, forest, X, dev, dtype=None):
        from isolation_forest_amd.ops import gpu_engine

        model = ExtendedIsolationForest(numEstimators=8,
                                        extensionLevel=0).fit(X[:600])
        model.forest = forest
        model._gpu_forest_cache = {}
        Xt = torch.from_numpy(X).to(dev)
        if dtype is not None:
            Xt = Xt.to(dtype)
        return gpu_engine.score_extended_forest(model, Xt, finalize=False)

    def test_f32_bitwise(self, dev):
        forest, X = self._forest()
        cpu_ps = cpu_engine.path_lengths_extended(forest, X)
        gpu_ps = self._score(forest, X, dev)
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32))

    def test_bf16_bitwise(self, dev):
        forest0, _ = self._forest(seed=42)
        rs = np.random.RandomState(43)
        Xb = torch.from_numpy(
            rs.normal(size=(4000, 6)).astype(np.float32)).to(torch.bfloat16)
        X = Xb.float().numpy()
        cpu_ps = cpu_engine.path_lengths_extended(forest0, X)
        from isolation_forest_amd.ops import gpu_engine

        model = ExtendedIsolationForest(numEstimators=8,
                                        extensionLevel=0).fit(X[:600])
        model.forest = forest0
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_extended_forest(
            model, Xb.to(dev), finalize=False)
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32))

    def test_foreign_weights_knife_edge_bitwise(self, dev):
        forest, X = self._forest(seed=44, foreign=True)
        cpu_ps = cpu_engine.path_lengths_extended(forest, X)
        gpu_ps = self._score(forest, X, dev)
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32))

    def test_matches_sparse_route(self, dev, monkeypatch):
        forest, X = self._forest(seed=45)
        s_eif0 = self._score(forest, X, dev).cpu().numpy()
        monkeypatch.setenv("IFA_EIF0_SPARSE", "1")
        s_sparse = self._score(forest, X, dev).cpu().numpy()
        np.testing.assert_array_equal(
            s_eif0.view(np.int32), s_sparse.view(np.int32))

    def test_nan_rows_route_to_strict_kernel(self, dev):
        """NaN features cannot take the mirrored-threshold walk (the
        oracle's NaN dot compares false at EVERY node); the route must
        detect them and stay bitwise via the strict-order kernel."""
        forest, X = self._forest(seed=47)
        rs = np.random.RandomState(48)
        X[rs.randint(0, len(X), 50), rs.randint(0, 6, 50)] = np.nan
        cpu_ps = cpu_engine.path_lengths_extended(forest, X)
        gpu_ps = self._score(forest, X, dev)
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32))

    def test_end_to_end_ext0_fit(self, dev):
        rs = np.random.RandomState(46)
        X = torch.from_numpy(
            rs.normal(size=(20000, 9)).astype(np.float32)).to(dev)
        model = ExtendedIsolationForest(
            numEstimators=100, extensionLevel=0, randomSeed=2).fit(X)
        s_gpu = model.score(X).cpu().numpy()
        s_cpu = model.score(X.float().cpu()).numpy()
        np.testing.assert_array_equal(
            s_gpu.view(np.int32), s_cpu.view(np.int32))


class TestForeignF64SplitsGPU:
    def test_knife_edge_foreign_model_bitwise(self, dev):
        """Foreign f64 splits strictly between adjacent f32s, rows planted
        ON the boundary: GPU ceil32 threshold keys must implement the
        exact f64 compare, bitwise vs the (f64-compare) CPU engine."""
        from isolation_forest_amd.ops import gpu_engine

        rs = np.random.RandomState(31)
        X = rs.normal(size=(2500, 4)).astype(np.float32)
        bag = cpu_engine.sample_bags(2500, 8, 128, seed=31, bootstrap=False)
        fs = cpu_engine.feature_subsets(4, 4, 8, seed=31)
        forest = cpu_engine.build_forest(X, bag, fs, 31, 128, 4, 4)
        internal = forest.feature >= 0
        v32 = forest.value.astype(np.float64)
        nxt = np.nextafter(forest.value,
                           np.float32(np.inf)).astype(np.float64)
        forest.value64 = np.where(internal, (v32 + nxt) / 2.0,
                                  forest.value64)
        ik = np.argwhere(internal)
        pick = ik[rs.randint(0, len(ik), size=600)]
        for r, (t, n) in enumerate(pick):
            X[r, forest.feature[t, n]] = forest.value[t, n]
        cpu_ps = cpu_engine.path_lengths(forest, X)
        model = IsolationForest(numEstimators=8).fit(X[:500])
        model.forest = forest
        model._gpu_forest_cache = {}
        gpu_ps = gpu_engine.score_forest(
            model, torch.from_numpy(X).to(dev), finalize=False)
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32))


class TestWideFormatFallbacks:
    """Beyond the packed-format caps (15-bit node ids => maxSamples 16384,
    12-bit feature ids => d 4094) the wide int4 kernels take over —
    VERDICT r01 #5: no config that fits can fail to score. Boundary on
    both sides, bitwise contracts."""

    def test_build_cap_boundary_16384_gpu_build(self, dev):
        X = torch.from_numpy(make_data(20000, 6, seed=70)).to(dev)
        model = IsolationForest(
            numEstimators=4, maxSamples=16384.0, randomSeed=19).fit(X)
        assert model.forest.num_samples == 16384
        s_gpu = model.score(X).cpu().numpy()
        s_cpu = model.score(X.float().cpu()).numpy()
        np.testing.assert_array_equal(
            s_gpu.view(np.int32), s_cpu.view(np.int32))

    def test_build_cap_boundary_16385_cpu_fallback(self, dev):
        """One past the GPU build cap: fit must still succeed on a CUDA
        tensor (CPU tree build + GPU wide scoring) and match the CPU
        model bitwise."""
        X = torch.from_numpy(make_data(20000, 6, seed=71)).to(dev)
        model = IsolationForest(
            numEstimators=2, maxSamples=16385.0, randomSeed=20).fit(X)
        assert model.forest.num_samples == 16385
        assert model.forest.feature.shape[1] > 32767  # wide territory
        s_gpu = model.score(X).cpu().numpy()
        s_cpu = model.score(X.float().cpu()).numpy()
        np.testing.assert_array_equal(
            s_gpu.view(np.int32), s_cpu.view(np.int32))

    def test_wide_d_4095_feature_ids(self, dev):
        """d = 4095 puts feature id 4094+ past the 12-bit packed field:
        routes to score_forest_wide, bitwise vs the oracle."""
        rs = np.random.RandomState(72)
        X = rs.normal(size=(2000, 4095)).astype(np.float32)
        bag = cpu_engine.sample_bags(2000, 3, 256, seed=21, bootstrap=False)
        fs = cpu_engine.feature_subsets(4095, 4095, 3, seed=21)
        forest = cpu_engine.build_forest(X, bag, fs, 21, 256, 4095, 4095)
        cpu_ps = cpu_engine.path_lengths(forest, X)
        model = IsolationForest(numEstimators=3).fit(X[:500])
        model.forest = forest
        model._gpu_forest_cache = {}
        from isolation_forest_amd.ops import gpu_engine

        assert not gpu_engine._packed_fits(forest, 4095)
        gpu_ps = gpu_engine.score_forest(
            model, torch.from_numpy(X).to(dev), finalize=False)
        np.testing.assert_array_equal(
            gpu_ps.cpu().numpy().view(np.int32), cpu_ps.view(np.int32))

    def test_extended_wide_nodes(self, dev):
        """EIF forest past 32767 nodes/tree: score_extended_wide keeps the
        strict j-order dot (bitwise)."""
        X = torch.from_numpy(make_data(25000, 5, seed=73)).to(dev)
        model = ExtendedIsolationForest(
            numEstimators=2, maxSamples=17000.0, randomSeed=22).fit(X)
        assert model.forest.feature.shape[1] > 32767
        s_gpu = model.score(X).cpu().numpy()
        s_cpu = model.score(X.float().cpu()).numpy()
        np.testing.assert_array_equal(
            s_gpu.view(np.int32), s_cpu.view(np.int32))


class TestParityFuzzSmoke:
    def test_ten_random_configs(self, dev):
        import subprocess
        import sys
        import os

        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        out = subprocess.run(
            [sys.executable, os.path.join(repo, "tools", "fuzz_parity.py"),
             "--iters", "10", "--seed", "42"],
            capture_output=True, text=True, timeout=600, cwd=repo)
        assert out.returncode == 0, out.stdout + out.stderr


class TestServingGPU:
    def test_http_scoring_runs_hip_kernels(self, dev, tmp_path, mammography):
        """The serving layer on a CUDA device must score through the same
        HIP path as batch transform (parity at float tolerance)."""
        from fastapi.testclient import TestClient

        from isolation_forest_amd.serving import create_app

        X, y = mammography
        model = IsolationForest(
            numEstimators=50, contamination=0.02, randomSeed=9
        ).fit(torch.from_numpy(X).to(dev))
        path = str(tmp_path / "m")
        model.save(path)
        client = TestClient(create_app(path, device=str(dev)))
        assert client.get("/v1/model").json()["device"] == str(dev)
        r = client.post("/v1/score", json={"instances": X[:256].tolist()})
        assert r.status_code == 200
        served = np.asarray(r.json()["scores"], dtype=np.float32)
        engine = model.score(torch.from_numpy(X[:256]).to(dev)).cpu().numpy()
        np.testing.assert_allclose(served, engine, rtol=0, atol=1e-6)
