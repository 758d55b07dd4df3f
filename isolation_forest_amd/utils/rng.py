"""Counter-based deterministic RNG (Philox4x32-10), numpy implementation.

Every random decision in the engine is a pure function of
``(randomSeed, purpose, treeId, index, attempt)`` — no RNG streams, no
order dependence. This gives the property the reference gets from
per-partition seeded ``scala.util.Random`` (SharedTrainLogic.scala:283-284,
BaggedPoint.scala:169-177) in a form that is identical on CPU (numpy),
GPU (HIP device code in ops/hip/philox.h) and any world size: the same
seed produces bit-identical forests on 1 CPU, 1 GPU or 8 GPUs.

The HIP device implementation in ops/hip/philox.h MUST stay in exact
correspondence with this file; tests/test_gpu.py checks forest-structure
equality between the two.
"""

from __future__ import annotations

import numpy as np

_M0 = np.uint32(0xD2511F53)
_M1 = np.uint32(0xCD9E8D57)
_W0 = np.uint32(0x9E3779B9)
_W1 = np.uint32(0xBB67AE85)

# Purpose tags — keep in sync with ops/hip/philox.h
P_BAG = 1  # row sampling for tree bags
P_FEATSUB = 2  # per-tree feature-subset permutation keys
P_FEATSEL = 3  # per-node split-feature Fisher-Yates draws
P_SPLIT = 4  # per-node split-point uniform
P_EIF_COORD = 5  # EIF per-node coordinate-subset Fisher-Yates draws
P_EIF_NORMAL = 6  # EIF Gaussian weight draws (Box-Muller)
P_EIF_INTERCEPT = 7  # EIF per-coordinate intercept uniforms
P_DATA = 8  # synthetic data generation (bench/tests)


def _mulhilo(a: np.uint32, b: np.ndarray):
    prod = a.astype(np.uint64) * b.astype(np.uint64)
    return (prod >> np.uint64(32)).astype(np.uint32), prod.astype(np.uint32)


def philox4x32(c0, c1, c2, c3, k0, k1):
    """Philox4x32-10. All args uint32 scalars/arrays (broadcastable).

    Returns 4 uint32 arrays.
    """
    c0 = np.asarray(c0, dtype=np.uint32)
    c1 = np.asarray(c1, dtype=np.uint32)
    c2 = np.asarray(c2, dtype=np.uint32)
    c3 = np.asarray(c3, dtype=np.uint32)
    c0, c1, c2, c3 = np.broadcast_arrays(c0, c1, c2, c3)
    c0, c1, c2, c3 = (x.copy() for x in (c0, c1, c2, c3))
    k0 = np.uint32(k0)
    k1 = np.uint32(k1)
    with np.errstate(over="ignore"):
        for _ in range(10):
            hi0, lo0 = _mulhilo(_M0, c0)
            hi1, lo1 = _mulhilo(_M1, c2)
            n0 = hi1 ^ c1 ^ k0
            n1 = lo1
            n2 = hi0 ^ c3 ^ k1
            n3 = lo0
            c0, c1, c2, c3 = n0, n1, n2, n3
            k0 = np.uint32((int(k0) + int(_W0)) & 0xFFFFFFFF)
            k1 = np.uint32((int(k1) + int(_W1)) & 0xFFFFFFFF)
    return c0, c1, c2, c3


def keys_from_seed(seed: int):
    """Split a 64-bit engine seed into the two Philox key words."""
    s = np.uint64(seed)
    return np.uint32(s & np.uint64(0xFFFFFFFF)), np.uint32(s >> np.uint64(32))


def u32(seed: int, purpose: int, tree, index, attempt=0):
    """First uint32 output for counter (attempt, index, tree, purpose)."""
    k0, k1 = keys_from_seed(seed)
    r0, _, _, _ = philox4x32(attempt, index, tree, np.uint32(purpose), k0, k1)
    return r0


def uniform(seed: int, purpose: int, tree, index, attempt=0):
    """float64 uniform in [0,1) with 24-bit resolution (matches HIP side)."""
    r = u32(seed, purpose, tree, index, attempt)
    return (r >> np.uint32(8)).astype(np.float64) * (1.0 / 16777216.0)


def uniform2(seed: int, purpose: int, tree, index, attempt=0):
    """Two independent uniforms (from output words 0 and 1)."""
    k0, k1 = keys_from_seed(seed)
    r0, r1, _, _ = philox4x32(attempt, index, tree, np.uint32(purpose), k0, k1)
    scale = 1.0 / 16777216.0
    return (
        (r0 >> np.uint32(8)).astype(np.float64) * scale,
        (r1 >> np.uint32(8)).astype(np.float64) * scale,
    )


def randint_below(seed: int, purpose: int, tree, index, bound, attempt=0):
    """Integer in [0, bound) via 32-bit multiply-shift (bound < 2^31).

    Slightly non-uniform for large bounds (standard fixed-point trick used
    identically on the HIP side: (u32 * bound) >> 32).
    """
    r = u32(seed, purpose, tree, index, attempt)
    bound_arr = np.asarray(bound, dtype=np.uint64)
    return ((r.astype(np.uint64) * bound_arr) >> np.uint64(32)).astype(np.int64)
