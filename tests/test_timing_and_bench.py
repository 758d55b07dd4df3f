"""Per-phase timing metrics + the driver's bench.py contract (CPU, gloo)."""

import json
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

from isolation_forest_amd import IsolationForest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class TestFitMetrics:
    def test_phases_recorded(self):
        rs = np.random.RandomState(0)
        X = rs.normal(size=(3000, 6)).astype(np.float32)
        model = IsolationForest(
            numEstimators=20, contamination=0.05, contaminationError=0.01
        ).fit(torch.from_numpy(X))
        m = model.fit_metrics
        for key in ["resolve", "bag+build", "gather", "threshold"]:
            assert key in m, key
            assert m[key] >= 0.0
        assert m.total > 0.0

    def test_no_threshold_phase_when_contamination_zero(self):
        rs = np.random.RandomState(1)
        X = rs.normal(size=(1000, 4)).astype(np.float32)
        model = IsolationForest(numEstimators=10).fit(torch.from_numpy(X))
        assert "threshold" not in model.fit_metrics


class TestBenchContract:
    """bench.py must print ONE valid JSON line on rank 0, single- and
    multi-process (the driver launches it via torch.distributed.run)."""

    def _check_payload(self, line, n_ranks):
        r = json.loads(line)
        assert r["unit"] == "rows/sec"
        assert r["higher_is_better"] is True
        assert r["scaling"] == "weak"
        assert r["steps"] >= 1 and r["value"] > 0
        assert r["config"]["rows_total"] == r["config"]["rows_per_gpu"] * n_ranks
        assert 0.5 < r["config"]["auroc"] <= 1.0
        return r

    def test_single_process(self):
        out = subprocess.run(
            [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
             "--rows", "4000", "--trees", "16"],
            cwd=REPO, capture_output=True, text=True, timeout=300,
        )
        assert out.returncode == 0, out.stderr[-2000:]
        lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
        assert len(lines) == 1
        self._check_payload(lines[0], 1)

    def test_two_process_gloo(self):
        env = dict(os.environ)
        env.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29711")
        out = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", "29711", "bench.py", "--gpus", "2",
             "--steps", "1", "--warmup", "0", "--rows", "4000",
             "--trees", "16"],
            cwd=REPO, capture_output=True, text=True, timeout=600, env=env,
        )
        assert out.returncode == 0, out.stderr[-2000:]
        lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
        assert len(lines) == 1, out.stdout
        r = self._check_payload(lines[0], 2)
        assert r["n_gpus"] in (0, 2)  # 0 = CPU plumbing mode
        assert r["config"]["world_size"] == 2

    def test_direct_gpus_flag_self_launches(self):
        """`python bench.py --gpus 2` invoked DIRECTLY must spawn 2 ranks
        (VERDICT.md round-1 gap #2: the driver runs exactly this)."""
        out = subprocess.run(
            [sys.executable, "bench.py", "--gpus", "2", "--steps", "1",
             "--warmup", "0", "--rows", "4000", "--trees", "16"],
            cwd=REPO, capture_output=True, text=True, timeout=600,
        )
        assert out.returncode == 0, out.stderr[-2000:]
        lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
        assert len(lines) == 1, out.stdout
        r = self._check_payload(lines[0], 2)
        assert r["config"]["world_size"] == 2
        assert r["config"]["rows_total"] == 8000


class TestTimingSync:
    def test_sync_env_flag(self, monkeypatch):
        from isolation_forest_amd.utils.timing import PhaseTimes, phase

        monkeypatch.setenv("IFA_TIMING_SYNC", "1")
        times = PhaseTimes()
        with phase(times, "work", device=torch.device("cpu")):
            pass
        assert "work" in times
        # None sink is a no-op
        with phase(None, "ignored"):
            pass

    def test_accumulates(self):
        from isolation_forest_amd.utils.timing import PhaseTimes, phase

        times = PhaseTimes()
        for _ in range(3):
            with phase(times, "p"):
                pass
        assert times["p"] >= 0.0
        assert times.total == sum(times.values())


class TestOfflineLint:
    def test_repo_is_lint_clean(self):
        out = subprocess.run(
            [sys.executable, "tools/lint_offline.py"],
            cwd=REPO, capture_output=True, text=True, timeout=120,
        )
        assert out.returncode == 0, out.stdout[-3000:]
