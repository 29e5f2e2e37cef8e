"""Custom-op layer: gfx950 HIP kernels with pure-torch CPU references.

Dispatch rule: CUDA (ROCm) tensors run the in-tree HIP extension
(``chinesener_amd/ops/_hip/*.so`` built by ``setup.py build_ext`` /
``scripts/build_ext.py``); CPU tensors run ``ops.reference``. On a GPU
box a missing extension is a LOUD error — no silent eager fallback —
unless CHINESENER_ALLOW_EAGER=1 (debug only).
"""
from __future__ import annotations

import os

import torch

_EXT = None
_EXT_ERR: Exception | None = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from . import _hip_ops  # in-tree built extension  # noqa: F401
        _EXT = _hip_ops
    except ImportError as e:
        _EXT_ERR = e
    return _EXT


def ext_available() -> bool:
    return _load_ext() is not None


def hip_enabled(t: torch.Tensor) -> bool:
    """True when this tensor should run the hand-written HIP path."""
    if not t.is_cuda:
        return False
    if ext_available():
        return True
    if os.environ.get("CHINESENER_ALLOW_EAGER") == "1":
        return False
    raise RuntimeError(
        "chinesener_amd HIP extension (_hip_ops) is not built but a GPU tensor "
        f"reached a custom op. Build it in-tree (python setup.py build_ext "
        f"--inplace with PYTORCH_ROCM_ARCH=gfx950). Import error: {_EXT_ERR}")


def get_ext():
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(f"_hip_ops extension missing: {_EXT_ERR}")
    return ext


from .functional import (  # noqa: E402,F401
    attention, attention_qkv, bias_gelu, add_layernorm, dropout_add_layernorm,
    embed3,
    layernorm, linear, crf_nll,
    crf_viterbi, bilstm, softlexicon_fuse, masked_cross_entropy, dice_loss,
    tener_attention,
)
