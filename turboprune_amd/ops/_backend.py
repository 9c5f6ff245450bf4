"""HIP extension loading & dispatch policy.

The compiled extension (``turboprune_amd/_C*.so``, built in-tree by
``setup.py build_ext --inplace`` for gfx950) provides the CDNA4 kernels.
Policy:

- On CPU tensors every op falls back to the plain PyTorch reference
  implementation (these are also the numerics oracles for the kernels).
- On GPU (ROCm) tensors the HIP kernels are REQUIRED: if the extension is
  missing we raise instead of silently degrading to eager torch — a GPU
  run must exercise the native path.
  (Set ``TURBOPRUNE_ALLOW_EAGER_GPU=1`` to permit the eager fallback when
  deliberately benchmarking against it.)
"""

from __future__ import annotations

import os

import torch

_EXT = None
_EXT_ERR: str = ""


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    try:
        from turboprune_amd import _C  # type: ignore
        _EXT = _C
    except ImportError as e:
        _EXT = None
        _EXT_ERR = str(e)
    return _EXT


def extension() :
    return _load_extension()


def has_extension() -> bool:
    return _load_extension() is not None


def allow_eager_gpu() -> bool:
    return os.environ.get("TURBOPRUNE_ALLOW_EAGER_GPU", "0") == "1"


def use_native(*tensors: torch.Tensor) -> bool:
    """True iff all tensors are on a ROCm device and the extension loaded.

    Raises if tensors are on GPU but the extension is absent (fail loudly:
    a GPU box must run the HIP path, not a silent eager fallback).
    """
    on_gpu = all(t.is_cuda for t in tensors if t is not None)
    if not on_gpu:
        return False
    if has_extension():
        return True
    if allow_eager_gpu():
        return False
    raise RuntimeError(
        "turboprune_amd HIP extension (turboprune_amd._C) is not built but "
        f"tensors are on GPU — refusing the silent eager fallback. "
        f"Build it with `python setup.py build_ext --inplace` "
        f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_EXT_ERR}"
    )
