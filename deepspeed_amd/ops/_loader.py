"""Loader for the in-tree compiled HIP extension (deepspeed_amd.ops._C).

The extension is built in-tree by ``setup.py build_ext --inplace`` (or
``__graft_entry__.build()``) for gfx950. Policy:
  * GPU present and extension missing -> hard error (never silently fall
    back to eager PyTorch on an MI355X box).
  * No GPU (CPU test host) -> ops fall back to reference torch
    implementations so the control-plane suite runs anywhere.
"""

import os

import torch

_C = None
_tried = False


def _try_import():
    global _C, _tried
    if _tried:
        return _C
    _tried = True
    if os.environ.get("DS_AMD_DISABLE_EXT") == "1":
        # debugging kill switch: force every op onto the torch fallback
        _C = None
        return None
    try:
        from . import _C as mod  # type: ignore
        _C = mod
    except ImportError as e:
        _C = None
        if torch.cuda.is_available() and os.environ.get("DS_AMD_ALLOW_EAGER") != "1":
            raise RuntimeError(
                "deepspeed_amd HIP extension (deepspeed_amd.ops._C) is not "
                "built but a GPU is present. Build it with "
                "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH="
                "gfx950). Set DS_AMD_ALLOW_EAGER=1 to force the slow eager "
                f"fallback. Import error: {e}") from e
    return _C


def get_ext():
    """Return the compiled module or None (CPU fallback allowed)."""
    return _try_import()


def has_ext() -> bool:
    return _try_import() is not None
