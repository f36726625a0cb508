"""Loader for the hand-written HIP (gfx950) extension.

The extension is built in-tree (``PYTORCH_ROCM_ARCH=gfx950 python setup.py
build_ext --inplace``) into ``distributed_embeddings_amd/_hip_ops*.so``.  On a GPU machine the HIP path is
mandatory: ops fail loudly if the extension is missing so a silent eager
fallback can never masquerade as the native path.  On CPU-only machines the
pure-PyTorch reference paths are used and the extension is not required.
"""

import importlib


_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _ext is not None or _tried:
        return _ext
    _tried = True
    try:
        _ext = importlib.import_module("distributed_embeddings_amd._hip_ops")
    except ImportError:
        _ext = None
    return _ext


def available() -> bool:
    return _load() is not None


def ops():
    """Returns the HIP extension module; raises if unavailable."""
    ext = _load()
    if ext is None:
        raise RuntimeError(
            "distributed_embeddings_amd._hip_ops extension is not built. "
            "Run `python setup.py build_ext --inplace` (requires hipcc; "
            "cross-compiles for gfx950 without a GPU)."
        )
    return ext
