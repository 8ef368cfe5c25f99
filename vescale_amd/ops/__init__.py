"""vescale_amd.ops — hand-written CDNA4 HIP kernels (L0).

The extension is built IN-TREE (`python setup.py build_ext --inplace`,
gfx950 only).  On a GPU box a missing extension is a HARD error — ops must
never fall back to eager silently (the round-end check records which .so
the GPU actually loaded).  On CPU-only boxes the reference implementations
in functional.py serve tests.
"""
from __future__ import annotations

import torch

_import_error = None
try:
    import importlib

    _C = importlib.import_module("._C", __name__)
except ImportError as e:  # pragma: no cover
    _C = None
    _import_error = e


def has_ext() -> bool:
    return _C is not None


def require_ext():
    if _C is None:
        if torch.cuda.is_available():
            raise RuntimeError(
                "vescale_amd HIP extension not built but a GPU is present. "
                "Run: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace\n"
                f"import error: {_import_error}"
            )
        raise RuntimeError("vescale_amd HIP extension unavailable (CPU-only box)")
    return _C


from .functional import (  # noqa: E402,F401
    flash_attention_causal,
    fused_cross_entropy,
    fused_add_rmsnorm,
    rmsnorm,
    rope_apply,
    rope_qkv,
    build_rope_table,
    swiglu,
    swiglu_packed,
    adamw_step_flat,
    l2norm_sq,
    scale_flat_,
    linear,
)
