"""Java/Scala-compatible shortest float formatting.

The reference's golden structural fingerprints (expectedTreeStructure.txt,
Nodes.scala:63-65 toString) embed Scala Double/Float toString output, which
follows Java's rules: shortest digits that round-trip, plain decimal for
1e-3 <= |x| < 1e7, otherwise ``d.dddE±e`` scientific notation, always at
least one digit after the decimal point.
"""

from __future__ import annotations

import numpy as np


def _java_fmt(x: float, is32: bool) -> str:
    if np.isnan(x):
        return "NaN"
    if np.isinf(x):
        return "Infinity" if x > 0 else "-Infinity"
    if x == 0.0:
        return "-0.0" if np.signbit(x) else "0.0"
    val = np.float32(x) if is32 else np.float64(x)
    m = abs(float(val))
    if 1e-3 <= m < 1e7:
        s = np.format_float_positional(val, unique=True)
        if s.endswith("."):
            s += "0"
        if "." not in s:
            s += ".0"
        return s
    s = np.format_float_scientific(val, unique=True)
    # numpy: '1.e-04' / '1.2345e+16' -> Java: '1.0E-4' / '1.2345E16'
    mant, exp = s.split("e")
    if mant.endswith("."):
        mant += "0"
    if "." not in mant:
        mant += ".0"
    e = int(exp)
    return f"{mant}E{e}"


def java_double(x: float) -> str:
    return _java_fmt(float(x), is32=False)


def java_float(x) -> str:
    return _java_fmt(float(np.float32(x)), is32=True)
