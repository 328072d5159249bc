"""Loader for the native HIP extension ``raft_amd._C``.

The extension is compiled in-tree for gfx950 (see setup.py / __graft_entry__.build).
On a GPU box the hot ops MUST run through it: if a CUDA(HIP) tensor reaches a hot
op and the extension is absent, we raise instead of silently falling back to a
slower eager path (the driver audits which .so files are actually loaded).
"""
from __future__ import annotations

import os

_EXT = None
_EXT_ERR: str | None = None


def _load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import torch  # noqa: F401  (must be imported first: torch symbols)
        from raft_amd import _C  # type: ignore

        _EXT = _C
    except Exception as e:  # pragma: no cover - exercised only when ext missing
        _EXT_ERR = f"{type(e).__name__}: {e}"
        _EXT = None
    return _EXT


def ext_or_none():
    """Return the extension module, or None when unavailable (CPU-only envs)."""
    return _load()


def require_ext():
    """Return the extension module; raise loudly when missing.

    Called on every GPU dispatch of a hot op. A GPU box without the built
    extension is a broken install — do not fall back silently.
    """
    ext = _load()
    if ext is None:
        raise RuntimeError(
            "raft_amd native extension (raft_amd/_C*.so) is not available "
            f"(import error: {_EXT_ERR}). Build it in-tree with "
            "`python setup.py build_ext --inplace` (hipcc, gfx950). "
            "GPU execution without the native kernels is disabled by design."
        )
    return ext


def has_ext() -> bool:
    return _load() is not None
