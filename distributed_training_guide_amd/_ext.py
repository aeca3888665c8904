"""Loader for the in-tree gfx950 HIP extension (_C.so).

Fail-loud policy: on a machine with a GPU, every op must run the HIP path;
a missing/unimportable extension raises instead of silently falling back to
eager PyTorch.  On CPU-only machines (the development container) the eager
fp32 references in ops/reference.py are used and the extension is optional.
"""
import os

import torch

_C = None
_load_error = None

try:
    from . import _C as _C_mod  # type: ignore[attr-defined]

    _C = _C_mod
except ImportError as e:  # extension not built (CPU-only dev is fine)
    _load_error = e


def has_ext() -> bool:
    return _C is not None


def ext():
    """Return the HIP extension module, raising loudly if unavailable."""
    if _C is None:
        raise RuntimeError(
            "distributed_training_guide_amd HIP extension (_C.so) is not "
            "built/importable but a GPU op was requested. Build it with "
            "`python tools/build_hip.py` (hipcc cross-compiles gfx950 "
            f"without a GPU). Original import error: {_load_error!r}"
        )
    return _C


def use_hip(*tensors) -> bool:
    """True iff these tensors live on the GPU (-> HIP kernels mandatory)."""
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if on_gpu and _C is None and os.environ.get("DTG_ALLOW_EAGER_GPU") != "1":
        ext()  # raises with the loud message
    return on_gpu and _C is not None
