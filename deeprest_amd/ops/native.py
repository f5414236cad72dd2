"""Loader for the in-tree HIP extension (deeprest_amd._C).

The extension is built ahead of time for gfx950 (setup.py / __graft_entry__
build()) and the resulting .so lives inside the package directory so it
travels with the repo snapshot to GPU machines — no JIT cache dependence.
"""

from __future__ import annotations

import importlib


_EXT = None
_TRIED = False


def load_native():
    """Import deeprest_amd._C if built; cache the result."""
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    try:
        import torch  # noqa: F401  (the .so links against torch's libs)

        _EXT = importlib.import_module("deeprest_amd._C")
    except ImportError:
        _EXT = None
    return _EXT


def native_available() -> bool:
    return load_native() is not None


def require_native(op_name: str):
    """Return the extension or raise — used on the GPU path only."""
    ext = load_native()
    if ext is None:
        raise RuntimeError(
            f"deeprest_amd._C is not built but op '{op_name}' was called on a GPU "
            f"tensor. Build the HIP extension first: `python setup.py build_ext "
            f"--inplace` (PYTORCH_ROCM_ARCH=gfx950). There is no GPU eager "
            f"fallback by design."
        )
    return ext
