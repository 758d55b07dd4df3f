"""Math + RNG + formatting unit tests (reference: core/UtilsTest.scala:9-17)."""

import numpy as np
import pytest

from isolation_forest_amd.utils.math import avg_path_length
from isolation_forest_amd.utils.javafmt import java_double, java_float
from isolation_forest_amd.utils import rng


class TestAvgPathLength:
    def test_golden_values(self):
        # goldens from core/UtilsTest.scala (float32 math)
        assert avg_path_length(0) == np.float32(0.0)
        assert avg_path_length(1) == np.float32(0.0)
        # c(2) = 2*(ln(1)+gamma) - 2*1/2 = 2*gamma - 1
        assert avg_path_length(2) == pytest.approx(2 * 0.5772156649 - 1.0, abs=1e-6)
        # the reference's Long.MaxValue golden
        assert avg_path_length(2**63 - 1) == np.float32(86.49098)

    def test_handbuilt_tree_goldens(self):
        # IsolationTreeTest.pathLengthTest: root split attr0 @1.5,
        # left leaf 10 instances, right leaf 20 instances
        assert np.float32(1.0 + avg_path_length(10)) == np.float32(4.7488804)
        assert np.float32(1.0 + avg_path_length(20)) == np.float32(6.143309)

    def test_vectorized(self):
        out = avg_path_length(np.array([0, 1, 2, 256]))
        assert out.dtype == np.float32
        assert out[0] == 0.0 and out[1] == 0.0
        assert out[3] > out[2]

    def test_monotone(self):
        vals = avg_path_length(np.arange(2, 10000))
        assert np.all(np.diff(vals) >= 0)


class TestJavaFmt:
    def test_double(self):
        assert java_double(4.0) == "4.0"
        assert java_double(0.5) == "0.5"
        assert java_double(-0.023960880394378714) == "-0.023960880394378714"
        assert java_double(0.8253754481933855) == "0.8253754481933855"
        assert java_double(1e-4) == "1.0E-4"
        assert java_double(1.5e7) == "1.5E7"
        assert java_double(0.0) == "0.0"

    def test_float(self):
        assert java_float(np.float32(0.3793424)) == "0.3793424"
        assert java_float(np.float32(-0.16987173)) == "-0.16987173"
        assert java_float(np.float32(1.0)) == "1.0"


class TestPhilox:
    def test_known_vector(self):
        # Philox4x32-10 reference test vector (counter=0, key=0) from the
        # Random123 known-answers set.
        r = rng.philox4x32(0, 0, 0, 0, 0, 0)
        assert [hex(int(x)) for x in r] == [
            "0x6627e8d5",
            "0xe169c58d",
            "0xbc57ac4c",
            "0x9b00dbd8",
        ]

    def test_determinism_and_broadcast(self):
        a = rng.u32(42, rng.P_SPLIT, np.arange(8)[:, None], np.arange(5)[None, :])
        b = rng.u32(42, rng.P_SPLIT, np.arange(8)[:, None], np.arange(5)[None, :])
        assert a.shape == (8, 5)
        assert np.array_equal(a, b)
        # different purposes decorrelate
        c = rng.u32(42, rng.P_BAG, np.arange(8)[:, None], np.arange(5)[None, :])
        assert not np.array_equal(a, c)

    def test_uniform_range_and_moments(self):
        u = rng.uniform(1, rng.P_SPLIT, 0, np.arange(200000))
        assert u.min() >= 0.0 and u.max() < 1.0
        assert abs(u.mean() - 0.5) < 0.005
        assert abs(u.var() - 1.0 / 12) < 0.003

    def test_randint_below(self):
        r = rng.randint_below(1, rng.P_BAG, 0, np.arange(100000), 7)
        assert r.min() >= 0 and r.max() <= 6
        counts = np.bincount(r, minlength=7)
        assert counts.min() > 0.9 * 100000 / 7


class TestBf16Round:
    def test_matches_torch_rne(self):
        import torch

        from isolation_forest_amd.utils.det_math import bf16_round

        rs = np.random.RandomState(0)
        a = rs.normal(size=50000).astype(np.float32)
        a *= rs.choice([1e-8, 1.0, 1e8], 50000).astype(np.float32)
        ours = bf16_round(a)
        theirs = torch.from_numpy(a).to(torch.bfloat16).float().numpy()
        np.testing.assert_array_equal(
            ours.view(np.int32), theirs.view(np.int32)
        )

    def test_ties_to_even_and_shape(self):
        from isolation_forest_amd.utils.det_math import bf16_round

        # 1.0 + 2^-9 is exactly halfway between bf16 neighbours 1.0 and
        # 1.0078125; RNE picks the even mantissa (1.0)
        tie = np.float32(1.0 + 2.0 ** -9)
        assert bf16_round(np.array([tie]))[0] == np.float32(1.0)
        out = bf16_round(np.zeros((3, 4), dtype=np.float32))
        assert out.shape == (3, 4)


class TestDetGaussianEdges:
    def test_u1_zero_is_finite_zero(self):
        """u1 == 0 (a 2^-24 Philox draw, fuzz-found) must give the true
        Box-Muller radial term 0, not sqrt(negative) = NaN."""
        from isolation_forest_amd.utils.det_math import det_gaussian

        u1 = np.array([0.0, 2.0 ** -24, 0.5, 1.0 - 2.0 ** -24, 1.0 - 1e-16])
        u2 = np.array([0.9186, 0.1, 0.2, 0.3, 0.77])
        g = det_gaussian(u1, u2)
        assert np.isfinite(g).all()
        assert g[0] == 0.0

    def test_grid_finite(self):
        from isolation_forest_amd.utils.det_math import det_gaussian

        rs = np.random.RandomState(0)
        u1 = np.concatenate([rs.rand(20000),
                             np.arange(16) * 2.0 ** -24,
                             1.0 - np.arange(1, 16) * 2.0 ** -24])
        u2 = rs.rand(len(u1))
        g = det_gaussian(u1, u2)
        assert np.isfinite(g).all()
