"""HIP extension loader.

``load()`` imports the in-tree gfx950 extension (_jsonproto.so).  On a GPU
box a missing/broken extension is a HARD error — there is deliberately no
silent eager fallback for the hot path (the CPU reference transcoder exists
only as the differential-test oracle and the explicit per-request fallback
for E_UNSUPPORTED slots, both counted in /metrics).
"""

from __future__ import annotations

import importlib
import sys
from pathlib import Path
from typing import Optional

_mod = None


class ExtensionUnavailable(RuntimeError):
    pass


def load(build_if_missing: bool = True):
    """Import (building if necessary) the _jsonproto extension."""
    global _mod
    if _mod is not None:
        return _mod
    so = Path(__file__).resolve().parent / "_jsonproto.so"
    if not so.exists() and build_if_missing:
        from . import build as build_mod

        try:
            build_mod.build()
        except Exception as e:
            raise ExtensionUnavailable(f"failed to build _jsonproto: {e}") from e
    if not so.exists():
        raise ExtensionUnavailable(f"{so} not built (run python -m ggrmcp_amd.ops.build)")
    try:
        from . import _jsonproto  # type: ignore
    except ImportError as e:
        raise ExtensionUnavailable(f"failed to import _jsonproto: {e}") from e
    _mod = _jsonproto
    return _mod


def gpu_available() -> bool:
    try:
        return load().device_count() > 0
    except ExtensionUnavailable:
        return False
