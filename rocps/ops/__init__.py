"""Hand-written HIP/CDNA4 ops (csrc/kernels) — capabilities C8–C11.

The extension ``rocps._hipops`` is built in-tree for gfx950 by
``setup.py`` / ``__graft_entry__.build()``. Policy:

- on CUDA(HIP) tensors the HIP kernels are THE path — if the extension
  is missing on a GPU machine, ops raise instead of silently falling
  back to eager PyTorch;
- on CPU tensors, ops run a plain PyTorch fp32 reference (the same
  reference the GPU numerics tests compare against).
"""

from __future__ import annotations

import torch

_ext = None
_ext_err: Exception = None


def _load():
    global _ext, _ext_err
    if _ext is None and _ext_err is None:
        try:
            from rocps import _hipops  # built in-tree
            _ext = _hipops
        except ImportError as e:
            _ext_err = e
    return _ext


def hip_ext():
    """The HIP extension module; raises loudly if unavailable."""
    ext = _load()
    if ext is None:
        raise RuntimeError(
            "rocps HIP kernels (rocps._hipops) are not built; on a GPU box "
            "this is a build error — run `python setup.py build_ext "
            f"--inplace` (PYTORCH_ROCM_ARCH=gfx950). Import error: {_ext_err}")
    return ext


def hip_available() -> bool:
    return _load() is not None


def require_hip_for(t: torch.Tensor) -> bool:
    """True -> dispatch to the HIP kernel; False -> CPU reference path."""
    if t.is_cuda:
        hip_ext()  # raises if missing: no silent eager fallback on GPU
        return True
    return False
