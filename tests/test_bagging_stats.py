"""Statistical validation of the bagging distributions.

Mirrors BaggedPointTest.scala:60-153, which asserts the mean/stddev of
Poisson (with-replacement) and Binomial (without-replacement) subsample
weights over seeds at eps = 0.01. Our bagging is fixed-n:

* bootstrap=False: a uniform n-subset without replacement per tree — the
  SAME distribution the reference realises end-to-end (Bernoulli(rate)
  inclusion followed by shuffle + slice(0, n) conditions on the count,
  giving a uniform n-subset; SharedTrainLogic.scala:287).
* bootstrap=True: fixed-n sampling WITH replacement (multinomial counts)
  — exactly the reference's Poisson bagging CONDITIONED on the bag
  totalling n (Poisson splitting property). The reference's shuffle+slice
  also conditions its oversampled Poisson pool to exactly n rows, so the
  realised bag-size distribution is identical (always n) and per-row
  multiplicities agree to O(1/N) (documented in PARITY.md).

The tests below quantify both: per-row inclusion frequencies, per-row
multiplicity mean/variance vs the Binomial(n, 1/N) marginal, duplicate
behaviour and seed reproducibility.
"""

import numpy as np

from isolation_forest_amd.core import cpu_engine


class TestWithoutReplacement:
    def test_exact_n_distinct_rows(self):
        bags = cpu_engine.sample_bags(1000, 50, 256, seed=1, bootstrap=False)
        assert bags.shape == (50, 256)
        for t in range(50):
            assert len(np.unique(bags[t])) == 256
            assert bags[t].min() >= 0 and bags[t].max() < 1000

    def test_inclusion_frequency_matches_rate(self):
        """Per-row inclusion probability must be n/N (mean) with the
        hypergeometric-style concentration a uniform subset implies —
        the analog of BaggedPointTest's Binomial mean/stddev check."""
        N, n, T = 500, 100, 400
        counts = np.zeros(N)
        for seed in range(8):
            bags = cpu_engine.sample_bags(N, T, n, seed=seed, bootstrap=False)
            hit = np.zeros(N)
            np.add.at(hit, bags.ravel(), 1)
            counts += hit
        trials = 8 * T
        freq = counts / trials
        rate = n / N
        # each row's inclusion is Bernoulli(rate) across trees/seeds:
        # mean within 4 sigma of rate, and the empirical std of the
        # per-row frequencies matches sqrt(rate*(1-rate)/trials)
        sigma = np.sqrt(rate * (1 - rate) / trials)
        assert np.abs(freq.mean() - rate) < 4 * sigma / np.sqrt(N)
        assert np.abs(freq.std() - sigma) < 0.2 * sigma

    def test_rejects_oversized(self):
        import pytest

        with pytest.raises(ValueError):
            cpu_engine.sample_bags(100, 2, 101, seed=1, bootstrap=False)


class TestWithReplacement:
    def test_multiplicity_moments_match_multinomial(self):
        """bootstrap=True: per-row multiplicity in a bag is the
        Multinomial(n, 1/N) marginal = Binomial(n, 1/N); check mean and
        variance over many (tree, seed) draws at the reference's
        eps = 0.01 spirit (BaggedPointTest.scala:60-112)."""
        N, n, T = 300, 256, 300
        mults = []
        for seed in range(6):
            bags = cpu_engine.sample_bags(N, T, n, seed=seed, bootstrap=True)
            for t in range(0, T, 10):
                m = np.zeros(N)
                np.add.at(m, bags[t], 1)
                mults.append(m)
        M = np.stack(mults)  # [draws, N]
        p = 1.0 / N
        mean_expect = n * p
        var_expect = n * p * (1 - p)
        assert np.abs(M.mean() - mean_expect) < 0.01 * mean_expect + 0.01
        assert np.abs(M.var() - var_expect) < 0.05 * var_expect

    def test_duplicates_present_and_bag_size_fixed(self):
        N, n = 300, 256
        bags = cpu_engine.sample_bags(N, 20, n, seed=3, bootstrap=True)
        assert bags.shape == (20, 256)
        # with n ~ N, duplicates are near-certain per bag
        dup_bags = sum(len(np.unique(bags[t])) < n for t in range(20))
        assert dup_bags == 20

    def test_oversized_bags_allowed(self):
        """With replacement, n > N is legal (the reference's Poisson
        bagging also produces them)."""
        bags = cpu_engine.sample_bags(50, 4, 200, seed=4, bootstrap=True)
        assert bags.shape == (4, 200)
        assert bags.max() < 50


class TestReproducibilityAndSharding:
    def test_same_seed_same_bags(self):
        a = cpu_engine.sample_bags(1000, 30, 128, seed=7, bootstrap=True)
        b = cpu_engine.sample_bags(1000, 30, 128, seed=7, bootstrap=True)
        np.testing.assert_array_equal(a, b)
        c = cpu_engine.sample_bags(1000, 30, 128, seed=8, bootstrap=True)
        assert not np.array_equal(a, c)

    def test_tree_offset_keys_global_ids(self):
        """Sharded draws with tree_id_offset reproduce the corresponding
        slice of the single-shard draw (BaggedPointTest.scala:289-333
        analog for our counter-keyed scheme)."""
        full = cpu_engine.sample_bags(1000, 16, 64, seed=9, bootstrap=False)
        hi = cpu_engine.sample_bags(1000, 8, 64, seed=9, bootstrap=False,
                                    tree_id_offset=8)
        np.testing.assert_array_equal(full[8:], hi)
