"""Isolation-forest math primitives.

Mirrors the semantics of the reference's core/Utils.scala:74-92: the average
path length c(n) of an unsuccessful BST search (Eq. 1 of Liu et al. 2008),
computed in float32 exactly as the Scala code does
(``2*(log(n-1)+gamma) - 2*(n-1)/n`` with float arithmetic, 0 for n <= 1).
"""

from __future__ import annotations

import numpy as np

# Euler-Mascheroni constant as float32, as in core/Utils.scala:74
EULER_CONSTANT = np.float32(0.5772156649)


def avg_path_length(n) -> np.ndarray | np.float32:
    """c(n): expected path length of an unsuccessful BST search over n points.

    Accepts a scalar or ndarray of counts (int or float). Returns float32,
    computed with float32 arithmetic to match the reference
    (core/Utils.scala:85-92) bit-for-bit on its golden values
    (e.g. c(2**63-1) == 86.49098f).
    """
    arr = np.asarray(n, dtype=np.float64)
    nf = arr.astype(np.float32)
    with np.errstate(divide="ignore", invalid="ignore"):
        ln = np.log(np.float32(nf - np.float32(1.0)), dtype=np.float32)
        term1 = np.float32(2.0) * (ln + EULER_CONSTANT)
        term2 = (np.float32(2.0) * (nf - np.float32(1.0)) / nf).astype(np.float32)
        out = (term1 - term2).astype(np.float32)
    out = np.where(arr <= 1.0, np.float32(0.0), out)
    if np.isscalar(n) or (isinstance(n, np.ndarray) and n.ndim == 0):
        return np.float32(out)
    return out.astype(np.float32)
