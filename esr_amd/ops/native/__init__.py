"""Loader for the in-tree gfx950 HIP extension (_esr_native).

The extension is built in-tree (esr_amd/ops/native/_esr_native.so) by
``__graft_entry__.build()`` / ``python -m esr_amd.ops.native.build`` so that
the built artifact travels with repo snapshots to GPU boxes.
"""

from __future__ import annotations

import os

_EXT = None
_TRIED = False


def get_ext():
    """Return the native extension module, or None if it is not built."""
    global _EXT, _TRIED
    if not _TRIED:
        _TRIED = True
        try:
            import torch  # noqa: F401  (extension links against torch libs)
            import importlib.util
            here = os.path.dirname(__file__)
            so = None
            for fn in os.listdir(here):
                if fn.startswith("_esr_native") and fn.endswith(".so"):
                    so = os.path.join(here, fn)
                    break
            if so is not None:
                spec = importlib.util.spec_from_file_location("_esr_native", so)
                mod = importlib.util.module_from_spec(spec)
                spec.loader.exec_module(mod)
                _EXT = mod
        except Exception:
            _EXT = None
    return _EXT


def native_available() -> bool:
    return get_ext() is not None


def require_ext():
    ext = get_ext()
    if ext is None:
        raise RuntimeError(
            "esr_amd native HIP extension is not built. Run "
            "`python -m esr_amd.ops.native.build` (or __graft_entry__.build()) "
            "before running GPU code - there is no eager GPU fallback.")
    return ext
