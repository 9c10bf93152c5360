"""Loader for the in-tree HIP extension (gfx950).

The extension is built IN-TREE (``python setup.py build_ext --inplace`` or
``python -m active_learning_amd.ops.build``) so the resulting .so travels with
the repo snapshot to GPU machines. It is never JIT-cached outside the tree.
"""

import importlib
import os

_C = None
_LOAD_ERROR = None


def load_extension():
    """Import the compiled extension module, caching the result."""
    global _C, _LOAD_ERROR
    if _C is not None:
        return _C
    try:
        import torch  # noqa: F401 — brings libtorch/libc10 into the process
        _C = importlib.import_module("active_learning_amd._C")
    except ImportError as e:  # keep the error for diagnostics
        _LOAD_ERROR = e
        _C = None
    return _C


def extension_available() -> bool:
    return load_extension() is not None


def require_extension():
    """Return the extension or raise loudly — called on every GPU-op dispatch."""
    ext = load_extension()
    if ext is None:
        raise RuntimeError(
            "active_learning_amd._C (the HIP/gfx950 kernel extension) is not built, "
            "but a GPU tensor reached the op layer. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original import error: {_LOAD_ERROR!r}. "
            "There is intentionally no torch fallback on GPU.")
    return ext


def _debug_fallback_enabled() -> bool:
    """Dev-only escape hatch; never set in CI or benchmarks."""
    return os.environ.get("AL_AMD_UNSAFE_GPU_TORCH_FALLBACK") == "1"
