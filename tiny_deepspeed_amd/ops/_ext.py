"""Loader for the in-tree HIP extension (_C).

Policy (driver contract): on a GPU box, ops MUST run the hand-written CDNA4
kernels — if the extension is missing we raise loudly instead of silently
falling back to eager PyTorch. On CPU-only machines (the build container) the
torch reference implementations are used and the extension is optional.
"""

import os

import torch

_EXT = None
_EXT_ERR = None
_TRIED = False


def _try_import():
    global _EXT, _EXT_ERR, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    try:
        from tiny_deepspeed_amd import _C  # built by `python setup.py build_ext --inplace`

        _EXT = _C
    except ImportError as e:  # pragma: no cover - exercised on GPU boxes
        _EXT = None
        _EXT_ERR = e
    return _EXT


def ext_available() -> bool:
    return _try_import() is not None


def get_ext():
    """Return the HIP extension module, or None on CPU-only hosts.

    Raises RuntimeError if a GPU is visible but the extension is not built:
    silent eager fallback on the GPU is forbidden.
    """
    ext = _try_import()
    if ext is None and torch.cuda.is_available() and not os.environ.get("TDSA_ALLOW_EAGER"):
        raise RuntimeError(
            "tiny_deepspeed_amd HIP extension (_C) is not built but a GPU is "
            f"visible. Build it in-tree with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_EXT_ERR}"
        )
    return ext


def use_native(*tensors) -> bool:
    """True when the op should dispatch to the HIP kernel."""
    if not tensors or not tensors[0].is_cuda:
        return False
    get_ext()  # raises loudly if on GPU without the extension
    return _EXT is not None
