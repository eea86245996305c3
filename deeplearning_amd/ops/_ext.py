"""HIP extension loader.

Policy (per framework design): on a machine with a GPU, the extension MUST be
present — ops raise rather than silently falling back to eager (so GPU tests
cannot pass on a PyTorch fallback). On CPU-only machines ops use their eager
reference implementations (used by the CPU test suite as numerics references).
"""
from __future__ import annotations

import torch

_EXT = None
_TRIED = False


def _load():
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    try:
        from . import _dla_hip  # built in-tree by setup.py build_ext --inplace

        _EXT = _dla_hip
    except ImportError as e:
        _EXT = None
        _IMPORT_ERR[0] = e
    return _EXT


_IMPORT_ERR = [None]


def has_ext() -> bool:
    return _load() is not None


def ext():
    """Return the extension module; raise with a clear message if missing."""
    m = _load()
    if m is None:
        raise RuntimeError(
            "deeplearning_amd HIP extension (_dla_hip) is not built. "
            "Run `PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            f"at the repo root. Original import error: {_IMPORT_ERR[0]}"
        )
    return m


def dense(t: torch.Tensor) -> torch.Tensor:
    """Return a densely-laid-out tensor WITHOUT destroying channels_last:
    .contiguous() on a channels_last tensor would copy to NCHW."""
    if t.is_contiguous():
        return t
    if t.dim() == 4 and t.is_contiguous(memory_format=torch.channels_last):
        return t
    return t.contiguous()


def same_layout(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Make b match a's memory format (for binary elementwise kernels)."""
    if a.dim() == 4 and a.is_contiguous(memory_format=torch.channels_last) \
            and not a.is_contiguous():
        return b.contiguous(memory_format=torch.channels_last)
    return b.contiguous()


import os

def _force_eager() -> bool:
    return os.environ.get("DLA_FORCE_EAGER", "0") == "1"  # A/B testing only


def use_hip(*tensors) -> bool:
    """True iff all tensors are on GPU. On GPU the extension is REQUIRED:
    if it is missing this raises instead of falling back."""
    if _force_eager():
        return False
    on_gpu = all(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_gpu:
        return False
    ext()  # raises if missing
    return True
