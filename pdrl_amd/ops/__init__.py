"""HIP/CDNA4 fused-op layer.

On a GPU box the compiled extension (``pdrl_amd/ops/_hip_ops*.so``, built
in-tree by ``setup.py build_ext --inplace`` / ``__graft_entry__.build()``)
MUST be present: ops fail loudly rather than silently falling back to eager
when CUDA/HIP is available. On CPU-only machines the eager paths are used and
the extension is optional.
"""
from __future__ import annotations

import torch

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        from pdrl_amd.ops import _hip_ops  # type: ignore

        _ext = _hip_ops
    except ImportError:
        _ext = None
        if torch.cuda.is_available():
            raise RuntimeError(
                "pdrl_amd HIP extension (_hip_ops) is not built but a GPU is "
                "present. Build it in-tree first: `python setup.py "
                "build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
            )
    return _ext


def available() -> bool:
    """True when the HIP extension is importable (GPU fused paths usable)."""
    if not torch.cuda.is_available():
        return False
    return _load() is not None


def ext():
    e = _load()
    if e is None:
        raise RuntimeError("pdrl_amd HIP extension not available")
    return e


def seq_lstm_forward(core, x, hx, cx, x2=None):
    """Fused body+LSTM+heads forward via the HIP kernel (autograd-capable).
    Dual-body cores (continuous critic) pass their second input as x2."""
    from .fused_core import seq_lstm_apply

    return seq_lstm_apply(core, x, hx, cx, x2)
