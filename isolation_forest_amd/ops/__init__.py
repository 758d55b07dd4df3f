"""HIP-extension loader.

The extension is built IN-TREE (``python setup.py build_ext --inplace`` or
``__graft_entry__.build()``) as ``isolation_forest_amd/ops/_iforest_hip.so``
for gfx950 only. On a GPU machine a missing extension is a HARD ERROR —
there is deliberately no silent eager/CPU fallback on the device path.
"""

from __future__ import annotations

import glob
import importlib
import os

_EXT = None
_EXT_ERR = None


def load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    here = os.path.dirname(__file__)
    sos = glob.glob(os.path.join(here, "_iforest_hip*.so"))
    try:
        import torch  # noqa: F401  (torch must be imported before the ext)

        if not sos:
            raise ImportError(
                f"no _iforest_hip*.so under {here}; build it with "
                "`python setup.py build_ext --inplace` (hipcc, gfx950)"
            )
        _EXT = importlib.import_module("isolation_forest_amd.ops._iforest_hip")
        return _EXT
    except Exception as e:  # pragma: no cover
        _EXT_ERR = e
        raise


def extension_available() -> bool:
    try:
        load_extension()
        return True
    except Exception:
        return False
