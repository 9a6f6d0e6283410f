"""Hand-written CDNA4 (gfx950) HIP kernels, exposed as autograd-capable ops.

The extension is built IN-TREE (setup.py build_ext --inplace, or
``python -m pipegoose_amd.ops.build``) so the .so ships with the repo snapshot
to the GPU box.  On a GPU box the HIP path is mandatory: ops raise if the
extension is missing rather than silently falling back to eager PyTorch.
CPU execution (tests, gloo plumbing) uses the eager reference path.
"""
import importlib
import os

_EXT = None
_EXT_ERR = None


def _load():
    global _EXT, _EXT_ERR
    if os.environ.get("PIPEGOOSE_DISABLE_EXT") == "1":  # debugging only
        return None
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        _EXT = importlib.import_module("pipegoose_amd.ops._C")
    except ImportError as e:
        _EXT_ERR = e
        _EXT = None
    return _EXT


def get_extension(required: bool = False):
    """Return the compiled HIP extension module (or None on CPU-only hosts)."""
    ext = _load()
    if ext is None and required:
        raise RuntimeError(
            "pipegoose_amd HIP extension (pipegoose_amd/ops/_C*.so) is not built "
            "but a GPU is present — build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {_EXT_ERR}"
        )
    return ext


def has_extension() -> bool:
    return _load() is not None
