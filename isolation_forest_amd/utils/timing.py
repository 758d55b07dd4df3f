"""Per-phase wall-clock timing for fit/transform.

The reference has no tracing at all; the engine-side
observability here is (a) these per-phase timers (resolve / bag+build /
gather / score / threshold), surfaced as ``model.fit_metrics`` and logged
at DEBUG, and (b) rocprofv3 kernel captures for the HIP kernels
(profiles/ in the repo root).

GPU phases are asynchronous: by default timers measure host wall time
without forcing a device sync (zero overhead in production). Set
``IFA_TIMING_SYNC=1`` to synchronize the device around every phase for
accurate per-phase GPU attribution (profiling runs only).
"""

from __future__ import annotations

import logging
import os
import time
from contextlib import contextmanager
from typing import Dict, Optional

import torch

logger = logging.getLogger(__name__)


def _sync_enabled() -> bool:
    return os.environ.get("IFA_TIMING_SYNC", "0") == "1"


class PhaseTimes(Dict[str, float]):
    """Phase-name -> seconds. A plain dict with a pretty logger."""

    def log(self, label: str, level: int = logging.DEBUG) -> None:
        if self:
            parts = ", ".join(f"{k}={v * 1e3:.1f}ms" for k, v in self.items())
            logger.log(level, "%s timings: %s", label, parts)

    @property
    def total(self) -> float:
        return sum(self.values())


@contextmanager
def phase(times: Optional[PhaseTimes], name: str, device=None):
    """Accumulate the wall time of the enclosed block into times[name].

    ``device``: a torch device (or tensor) whose CUDA stream is synced
    before/after the block when IFA_TIMING_SYNC=1.
    """
    if times is None:
        yield
        return
    sync = _sync_enabled()
    is_cuda = device is not None and (
        device.is_cuda if isinstance(device, torch.Tensor)
        else getattr(device, "type", None) == "cuda"
    )
    if sync and is_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    try:
        yield
    finally:
        if sync and is_cuda:
            torch.cuda.synchronize()
        times[name] = times.get(name, 0.0) + (time.perf_counter() - t0)
