"""Deterministic transcendental functions (bitwise CPU == GPU).

The EIF hyperplane weights are Gaussian draws via Box-Muller, which needs
log and cos. libm implementations differ between numpy/glibc and HIP
device code by ULPs, which would (very rarely, but nonzero) flip a
float32-rounded weight and break the bitwise CPU<->GPU forest-structure
parity the test suite enforces. These fixed Horner polynomials use only
IEEE-correctly-rounded double ops (+,-,*,/ and sqrt) in a fixed order, so
numpy and the HIP device implementation (ops/hip/det_math.h) produce
BIT-IDENTICAL doubles. Accuracy ~1e-16 relative — far beyond the float32
precision the weights are stored at.
"""

from __future__ import annotations

import numpy as np

_LN2 = 6.93147180559945286227e-01  # double closest to ln 2

# atanh series coefficients 2/(2k+1), k=0..10
_LOG_COEFFS = [2.0 / (2 * k + 1) for k in range(10, -1, -1)]

def _factorial(n):
    out = 1
    for i in range(2, n + 1):
        out *= i
    return out


_COS_C = [((-1.0) ** k) / _factorial(2 * k) for k in range(8, -1, -1)]
_SIN_C = [((-1.0) ** k) / _factorial(2 * k + 1) for k in range(8, -1, -1)]


def det_log(x):
    """Natural log for x in (0, 1]; vectorized; bitwise-deterministic."""
    x = np.asarray(x, dtype=np.float64)
    m, e = np.frexp(x)  # x = m * 2^e, m in [0.5, 1)
    z = (m - 1.0) / (m + 1.0)
    z2 = z * z
    s = np.full_like(z, _LOG_COEFFS[0])
    for c in _LOG_COEFFS[1:]:
        s = s * z2 + c
    s = s * z
    return s + e.astype(np.float64) * _LN2


def _poly(x2, coeffs):
    s = np.full_like(x2, coeffs[0])
    for c in coeffs[1:]:
        s = s * x2 + c
    return s


_PI_2 = 1.57079632679489661923


def det_cos2pi(u):
    """cos(2*pi*u) for u in [0,1); vectorized; bitwise-deterministic."""
    u = np.asarray(u, dtype=np.float64)
    s4 = u * 4.0
    q = np.floor(s4)
    f = s4 - q  # [0, 1): fraction of a quadrant
    qi = q.astype(np.int64) & 3
    # within quadrant: angle a = f * pi/2; reduce to <= pi/4 by mirroring
    use_sin_half = f > 0.5
    g = np.where(use_sin_half, 1.0 - f, f)
    a = g * _PI_2
    a2 = a * a
    cosv = _poly(a2, _COS_C)
    sinv = _poly(a2, _SIN_C) * a
    # cos(f*pi/2) = cos(a) if f<=0.5 else sin((1-f)*pi/2)
    cos_q = np.where(use_sin_half, sinv, cosv)
    # sin(f*pi/2) = sin(a) if f<=0.5 else cos((1-f)*pi/2)
    sin_q = np.where(use_sin_half, cosv, sinv)
    out = np.where(
        qi == 0, cos_q, np.where(qi == 1, -sin_q, np.where(qi == 2, -cos_q, sin_q))
    )
    return out


def bf16_round(a: np.ndarray) -> np.ndarray:
    """Round float32 values to bf16 (round-to-nearest-even), returned as
    float32 — numpy mirror of torch's .to(torch.bfloat16) and of the GPU
    dense-v3 weight staging (ops/hip/forest_kernels.hip). Finite inputs
    only (hyperplane weights are finite by construction)."""
    b = np.ascontiguousarray(a, dtype=np.float32).view(np.uint32)
    r = ((b >> np.uint32(16)) & np.uint32(1)) + np.uint32(0x7FFF)
    return ((b + r) & np.uint32(0xFFFF0000)).view(np.float32).reshape(
        np.asarray(a).shape)


def det_gaussian(u1, u2):
    """Box-Muller with deterministic log/cos; returns float64.

    det_log(1.0) carries a ~1e-12 polynomial residual rather than exact 0,
    so the u1 == 0 draw (probability 2^-24) would take sqrt of a negative
    and poison a hyperplane with NaN weights (fuzz-found). The true radial
    term at u1 = 0 is 0: clamp the sqrt argument at 0 (mirrored in
    ops/hip/det_math.h — bitwise CPU == GPU)."""
    u1 = np.asarray(u1, dtype=np.float64)
    a = -2.0 * det_log(np.maximum(1.0 - u1, 1e-300))
    r = np.where(u1 < 1.0, np.sqrt(np.maximum(a, 0.0)), 0.0)
    return r * det_cos2pi(u2)
