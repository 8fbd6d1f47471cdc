"""HIP extension loading.

The CDNA4 kernels live in ``sheeprl_amd/ops/csrc`` and are built IN-TREE
(``python setup.py build_ext --inplace`` or ``__graft_entry__.build()``) into
``sheeprl_amd/ops/_sheep_hip*.so`` so the binary travels with repo snapshots.

Policy: on a CUDA/ROCm device the HIP kernels are the only compute path — if
the extension is missing, ops raise instead of silently falling back to eager
PyTorch (the CPU fallback exists for CPU tensors / CPU tests only).
"""

from __future__ import annotations

import importlib
import os
from typing import Any, Optional

import torch

_EXT: Optional[Any] = None
_TRIED = False
_ERR: Optional[str] = None


def _try_load() -> None:
    global _EXT, _TRIED, _ERR
    if _TRIED:
        return
    _TRIED = True
    try:
        _EXT = importlib.import_module("sheeprl_amd.ops._sheep_hip")
    except Exception as e:  # noqa: BLE001
        _EXT = None
        _ERR = f"{type(e).__name__}: {e}"


def get_ext() -> Optional[Any]:
    _try_load()
    return _EXT


def has_ext() -> bool:
    return get_ext() is not None


def require_ext() -> Any:
    ext = get_ext()
    if ext is None:
        raise RuntimeError(
            "sheeprl_amd HIP extension (_sheep_hip) is not built but a GPU tensor "
            f"reached a fused op. Build it in-tree with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_ERR}"
        )
    return ext


def use_hip(*tensors: torch.Tensor) -> bool:
    """True when the computation should run on the HIP kernels."""
    if not tensors or not tensors[0].is_cuda:
        return False
    if os.environ.get("SHEEPRL_AMD_DISABLE_EXT", "0") == "1":
        return False
    return True
