"""Native data-plane bindings.

``load()`` imports the in-tree _hipshuffle extension. On a GPU box the
extension is REQUIRED — ops fail loudly rather than falling back to eager
paths silently (the HIP path must be the one that runs).
"""

from __future__ import annotations

import importlib

_mod = None


def load(build_if_missing: bool = True):
    global _mod
    if _mod is not None:
        return _mod
    try:
        from sparkrdma_amd.ops import _hipshuffle as mod  # type: ignore
    except ImportError:
        if not build_if_missing:
            raise
        from . import build as _b
        _b.ensure_built(verbose=True)
        importlib.invalidate_caches()
        from sparkrdma_amd.ops import _hipshuffle as mod  # type: ignore
    _mod = mod
    return mod


def available() -> bool:
    try:
        load(build_if_missing=False)
        return True
    except ImportError:
        return False
