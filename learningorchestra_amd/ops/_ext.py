"""Loader for the in-tree gfx950 kernel extension (_lo_C.so).

Policy (per the build contract): on a GPU host the HIP path MUST run — ops
fail loudly if the extension is missing; on CPU-only hosts (the CI container)
the torch reference implementations in ``functional.py`` are used instead.
"""
from __future__ import annotations

import importlib.util
import os
import sys
from typing import Optional

_PKG_DIR = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_SO_PATH = os.path.join(_PKG_DIR, "_build", "_lo_C.so")

_ext = None
_tried = False


def _load() -> Optional[object]:
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    if os.path.exists(_SO_PATH):
        spec = importlib.util.spec_from_file_location("_lo_C", _SO_PATH)
        mod = importlib.util.module_from_spec(spec)
        try:
            spec.loader.exec_module(mod)
            sys.modules["_lo_C"] = mod
            _ext = mod
            return _ext
        except Exception as exc:  # pragma: no cover - load failure surfaces below
            _load_error[0] = exc
            return None
    return None


_load_error = [None]


def ext() -> Optional[object]:
    return _load()


def has_ext() -> bool:
    return _load() is not None


def require_ext():
    """GPU code paths call this: loud failure, never a silent eager fallback."""
    mod = _load()
    if mod is None:
        raise RuntimeError(
            f"learningorchestra_amd HIP extension not found at {_SO_PATH} "
            f"(load error: {_load_error[0]!r}). Build it with "
            "`python -m learningorchestra_amd.build_ext` — GPU execution "
            "refuses to fall back to eager PyTorch.")
    return mod
