"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, PyTorch reference on CPU.

On a GPU box the in-tree extension `hypha_amd/_C*.so` MUST be importable —
ops raise loudly if it is missing (no silent eager fallback), so a GPU test
run always exercises the native kernels. On CPU (CI container) the reference
implementations run instead.

Set HYPHA_FORCE_REF=1 to force the reference path even on GPU (debug only).
"""

from __future__ import annotations

import os

import torch

from . import reference

_C = None
_import_error: Exception | None = None
try:  # built in-tree by setup.py / __graft_entry__.build()
    from hypha_amd import _C as _C  # type: ignore
except ImportError as e:  # pragma: no cover - exercised only on GPU boxes
    _import_error = e


def _force_ref() -> bool:
    return os.environ.get("HYPHA_FORCE_REF", "0") == "1"


def has_native() -> bool:
    return _C is not None


def native_available_or_raise() -> None:
    if _C is None:
        raise RuntimeError(
            "hypha_amd native HIP extension (hypha_amd/_C) is not built but a GPU "
            "is visible. Build it with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_import_error!r}"
        )


def use_native(*tensors: torch.Tensor) -> bool:
    """True when the op should run the HIP kernel path."""
    if not tensors or not tensors[0].is_cuda:
        return False
    if _force_ref():
        return False
    native_available_or_raise()
    return True


from .interface import (  # noqa: E402
    add_rmsnorm,
    apply_rope_qk,
    attn_decode,
    cross_entropy_loss,
    extract_delta,
    flash_attention,
    fused_adamw,
    fused_nesterov,
    gelu,
    layernorm,
    linear_skinny,
    rmsnorm,
    swiglu,
)

__all__ = [
    "has_native",
    "native_available_or_raise",
    "use_native",
    "reference",
    "rmsnorm",
    "layernorm",
    "gelu",
    "add_rmsnorm",
    "apply_rope_qk",
    "attn_decode",
    "swiglu",
    "flash_attention",
    "cross_entropy_loss",
    "fused_adamw",
    "fused_nesterov",
    "linear_skinny",
    "extract_delta",
]
