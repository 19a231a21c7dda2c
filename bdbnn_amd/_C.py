"""Loader for the in-tree HIP extension (bdbnn_amd/_native*.so).

The extension is built for gfx950 only (see csrc/ and setup.py:
``python setup.py build_ext --inplace``).  On a GPU machine the native
path is mandatory: if a CUDA tensor reaches an op and the extension is
missing, we raise instead of silently falling back to eager PyTorch.
On CPU-only machines (CI) the pure-PyTorch oracle path is used.
"""

import importlib

_native = None
_native_err = None


def _load():
    global _native, _native_err
    if _native is not None or _native_err is not None:
        return _native
    try:
        _native = importlib.import_module("bdbnn_amd._native")
    except ImportError as e:  # not built (CPU CI) or build mismatch
        _native_err = e
        _native = None
    return _native


def native():
    """Return the native module or None (CPU fallback allowed)."""
    return _load()


def native_required():
    """Return the native module; raise if missing.

    Called on the GPU hot path so a missing/broken extension fails loudly
    instead of silently running eager PyTorch on the GPU.
    """
    m = _load()
    if m is None:
        raise RuntimeError(
            "bdbnn_amd native HIP extension is not available on a GPU path. "
            "Build it in-tree with `python setup.py build_ext --inplace` "
            f"(original import error: {_native_err})"
        )
    return m


def has_native() -> bool:
    return _load() is not None
