"""Multi-process distributed tests (gloo backend, CPU, world_size 2).

These exercise the exact code paths the 8-GPU RCCL run uses (the Comm
wrapper is backend-agnostic); the reference's analog is the local[4]
SparkSession tests (core/TestUtils.scala:39-56).
"""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from tests.conftest import auroc

WORLD = 2


def _run_workers(fn, world=WORLD, args=()):
    ctx = mp.get_context("spawn")
    port = int(os.environ.get("TEST_DIST_PORT", "29541")) + (os.getpid() % 500)
    procs = []
    q = ctx.Queue()
    for rank in range(world):
        p = ctx.Process(target=_worker_entry, args=(fn, rank, world, port, q, args))
        p.start()
        procs.append(p)
    results = {}
    for _ in range(world):
        rank, payload = q.get(timeout=300)
        if isinstance(payload, str) and payload.startswith("ERROR:"):
            for p in procs:
                p.terminate()
            raise AssertionError(f"rank {rank} failed: {payload}")
        results[rank] = payload
    for p in procs:
        p.join(timeout=60)
    return results


def _worker_entry(fn, rank, world, port, q, args):
    import traceback

    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["LOCAL_RANK"] = str(rank)
        import torch.distributed as dist

        from isolation_forest_amd.parallel import init_from_env

        comm = init_from_env(backend="gloo")
        out = fn(comm, rank, world, *args)
        q.put((rank, out))
        dist.barrier()
        dist.destroy_process_group()
    except Exception:
        q.put((rank, "ERROR:" + traceback.format_exc()))


# ---------------------------------------------------------------------------
# worker bodies (module-level for spawn picklability)
# ---------------------------------------------------------------------------


def _body_collectives(comm, rank, world):
    assert comm.world_size == world
    total = comm.all_reduce_sum_int(10 + rank)
    mn = comm.all_reduce_min(float(rank))
    mx = comm.all_reduce_max(float(rank))
    t = torch.ones(4) * (rank + 1)
    comm.all_reduce(t)
    gathered = comm.all_gather_1d(torch.arange(rank + 1, dtype=torch.float32))
    return {
        "sum": total,
        "min": mn,
        "max": mx,
        "allreduce": t.tolist(),
        "gather": gathered.tolist(),
    }


def _body_tree_sharded_fit(comm, rank, world):
    """Every rank holds the SAME rows; the forest must be bit-identical to a
    single-process fit (tree draws are keyed by global tree id)."""
    from isolation_forest_amd import IsolationForest

    rs = np.random.RandomState(0)
    X = rs.normal(size=(3000, 5)).astype(np.float32)
    model = IsolationForest(numEstimators=11, randomSeed=5, contamination=0.05,
                            contaminationError=0.0).fit(X, comm=comm)
    fingerprints = [model.forest.tree_to_string(t) for t in range(11)]
    scores = model.score(torch.from_numpy(X[:100])).numpy()
    return {
        "fp": fingerprints,
        "scores": scores.tolist(),
        "threshold": model.outlier_score_threshold,
    }


def _body_row_sharded_fit(comm, rank, world):
    """Each rank holds a disjoint row shard (the 1B-row mode): bags are
    rank-local, the forest is all-gathered, threshold is global."""
    from isolation_forest_amd import IsolationForest

    rs = np.random.RandomState(100 + rank)
    d = 6
    inliers = rs.normal(size=(4000, d)).astype(np.float32)
    outliers = rs.uniform(-10, 10, size=(80, d)).astype(np.float32)
    X = np.concatenate([inliers, outliers])
    y = np.concatenate([np.zeros(4000), np.ones(80)])
    model = IsolationForest(
        numEstimators=40, randomSeed=3, contamination=0.02, contaminationError=0.0
    ).fit(X, comm=comm)
    out = model.transform(X)
    a = auroc(y, out["outlierScore"].numpy())
    flagged = float(out["predictedLabel"].sum())
    count = len(X)
    # global observed contamination via the comm
    tot = comm.all_reduce(torch.tensor([flagged, float(count)], dtype=torch.float64))
    return {
        "auroc": a,
        "observed": float(tot[0] / tot[1]),
        "trees": model.forest.num_trees,
        "threshold": model.outlier_score_threshold,
    }


def _body_extended_sharded(comm, rank, world):
    from isolation_forest_amd import ExtendedIsolationForest

    rs = np.random.RandomState(0)
    X = rs.normal(size=(2000, 4)).astype(np.float32)
    model = ExtendedIsolationForest(numEstimators=7, randomSeed=2).fit(X, comm=comm)
    return {
        "fp": [model.forest.tree_to_string(t) for t in range(7)],
        "ext": model.extension_level,
    }


# ---------------------------------------------------------------------------
# tests
# ---------------------------------------------------------------------------


class TestCollectives:
    def test_basic(self):
        res = _run_workers(_body_collectives)
        for rank in range(WORLD):
            assert res[rank]["sum"] == sum(10 + r for r in range(WORLD))
            assert res[rank]["min"] == 0.0
            assert res[rank]["max"] == WORLD - 1
            assert res[rank]["allreduce"] == [float(sum(range(1, WORLD + 1)))] * 4
            expect = [float(i) for r in range(WORLD) for i in range(r + 1)]
            assert res[rank]["gather"] == expect


class TestTreeShardedFit:
    def test_matches_single_process(self):
        res = _run_workers(_body_tree_sharded_fit)
        # both ranks agree exactly
        assert res[0]["fp"] == res[1]["fp"]
        assert res[0]["scores"] == res[1]["scores"]
        assert res[0]["threshold"] == res[1]["threshold"]
        # and match a single-process fit bit-for-bit
        from isolation_forest_amd import IsolationForest

        rs = np.random.RandomState(0)
        X = rs.normal(size=(3000, 5)).astype(np.float32)
        solo = IsolationForest(numEstimators=11, randomSeed=5, contamination=0.05,
                               contaminationError=0.0).fit(X)
        solo_fp = [solo.forest.tree_to_string(t) for t in range(11)]
        assert res[0]["fp"] == solo_fp
        assert res[0]["threshold"] == solo.outlier_score_threshold


class TestRowShardedFit:
    def test_quality_and_contamination(self):
        res = _run_workers(_body_row_sharded_fit)
        for rank in range(WORLD):
            assert res[rank]["trees"] == 40  # full forest gathered everywhere
            assert res[rank]["auroc"] > 0.9
            assert res[rank]["observed"] == pytest.approx(0.02, abs=0.01)
        assert res[0]["threshold"] == res[1]["threshold"]


class TestExtendedSharded:
    def test_matches_across_ranks(self):
        res = _run_workers(_body_extended_sharded)
        assert res[0]["fp"] == res[1]["fp"]
        assert res[0]["ext"] == 3


class TestWorldFour:
    """The reference's distributed tests run on a real local[4] session
    (core/TestUtils.scala:39-56); world-4 gloo is our analog of that
    parallelism degree — same collectives, uneven 11/4 tree shards."""

    def test_tree_sharded_world4_bitwise(self):
        res = _run_workers(_body_tree_sharded_fit, world=4)
        for r in range(1, 4):
            assert res[0]["fp"] == res[r]["fp"]
            assert res[0]["threshold"] == res[r]["threshold"]
        from isolation_forest_amd import IsolationForest

        rs = np.random.RandomState(0)
        X = rs.normal(size=(3000, 5)).astype(np.float32)
        solo = IsolationForest(numEstimators=11, randomSeed=5,
                               contamination=0.05,
                               contaminationError=0.0).fit(X)
        assert res[0]["fp"] == [solo.forest.tree_to_string(t)
                                for t in range(11)]
        assert res[0]["threshold"] == solo.outlier_score_threshold

    def test_row_sharded_world4(self):
        res = _run_workers(_body_row_sharded_fit, world=4)
        for r in range(4):
            assert res[r]["auroc"] > 0.85
            assert res[r]["trees"] == 40
            assert res[r]["observed"] == pytest.approx(0.02, abs=0.01)
        assert len({res[r]["threshold"] for r in range(4)}) == 1
