"""HIP extension loader.

The extension (``_mcdp_C``) is built IN-TREE by ``setup.py build_ext --inplace``
(or ``__graft_entry__.build()``) with ``PYTORCH_ROCM_ARCH=gfx950`` so the
built .so travels with the repo snapshot to GPU boxes.

Policy: on a ROCm GPU the HIP kernels are the ONLY compute path — if the
extension is missing we raise instead of silently falling back to eager
PyTorch. On CPU (no GPU present) ops use their pure-torch reference
implementations, which double as the numerics references in tests.
"""
from __future__ import annotations

import importlib
import os
from typing import Optional

import torch

_ext = None
_tried = False
_err: Optional[BaseException] = None


def _load():
    global _ext, _tried, _err
    if _tried:
        return _ext
    _tried = True
    try:
        _ext = importlib.import_module("mlx_cuda_distributed_pretraining_amd._mcdp_C")
    except Exception as e:  # extension not built
        _ext = None
        _err = e
    return _ext


def get_ext():
    """Return the extension module or None (CPU-only environments)."""
    return _load()


def require_ext():
    """Return the extension module; raise loudly if missing on a GPU box."""
    ext = _load()
    if ext is None:
        raise RuntimeError(
            "mlx_cuda_distributed_pretraining_amd HIP extension (_mcdp_C) is not built. "
            "On a GPU box this is a hard error: the HIP kernels are the compute path. "
            "Build it in-tree with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Original import error: {_err!r}"
        )
    return ext


def use_hip(*tensors: torch.Tensor,
            dtypes=(torch.bfloat16, torch.float32)) -> bool:
    """True when the op should dispatch to the HIP kernel path.

    ``dtypes``: dtypes the kernel supports — anything else (e.g. fp16 under
    ``system.precision: float16``) runs the torch reference composition,
    which is correct on GPU, just not the tuned path."""
    if not tensors:
        return False
    if not all(t.is_cuda for t in tensors if isinstance(t, torch.Tensor)):
        return False
    if any(t.dtype not in dtypes for t in tensors
           if isinstance(t, torch.Tensor) and t.is_floating_point()):
        return False
    if os.environ.get("MCDP_FORCE_TORCH") == "1":
        return False
    # On a GPU, the extension is mandatory: require_ext raises if missing.
    require_ext()
    return True
