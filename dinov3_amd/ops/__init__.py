"""MI355X op library.

Every hot op in the DINOv3 step has two implementations:

- a hand-written CDNA4 HIP kernel (``csrc/*.hip``, compiled for gfx950 into the
  in-tree extension ``dinov3_amd/ops/_hip_ops*.so``) — the path that runs on a
  GPU. On a CUDA/HIP device these ops REFUSE to fall back silently: if the
  extension is missing the op raises, so a GPU run can never silently use
  eager PyTorch.
- a plain PyTorch fp32-reference implementation — used on CPU (tests, gloo
  plumbing runs) and as the numerics oracle for the kernel unit tests.

Kernel inventory mirrors SURVEY.md §2.4 (K1-K28).
"""

from __future__ import annotations

import os

import torch

from .layernorm import layer_norm, rms_norm
from .fmha import fmha
from .rope import rope_apply
from .l2norm import l2_normalize
from .bias_act import bias_gelu
from .fused_update import ema_update_, multi_tensor_adamw_, grad_l2_norm_sq
from .proto_scores import dino_softmax_ce, ibot_softmax_ce, sinkhorn_knopp
from .ls_axpy import ls_axpy
from .row_ops import gather_rows, scatter_add_rows
from .flat_attention import flat_multi_fmha

__all__ = [
    "layer_norm",
    "rms_norm",
    "fmha",
    "rope_apply",
    "l2_normalize",
    "bias_gelu",
    "ema_update_",
    "multi_tensor_adamw_",
    "grad_l2_norm_sq",
    "dino_softmax_ce",
    "ibot_softmax_ce",
    "sinkhorn_knopp",
    "ls_axpy",
    "gather_rows",
    "scatter_add_rows",
    "flat_multi_fmha",
    "hip_ops",
    "has_hip_ops",
]

_hip_mod = None
_hip_tried = False


def _load_hip():
    global _hip_mod, _hip_tried
    if _hip_tried:
        return _hip_mod
    _hip_tried = True
    try:
        from . import _hip_ops  # built in-tree by setup.py / __graft_entry__.build()

        _hip_mod = _hip_ops
    except ImportError:
        try:
            import importlib

            _hip_mod = importlib.import_module("dinov3_amd.ops._hip_ops")
        except ImportError:
            _hip_mod = None
    return _hip_mod


def has_hip_ops() -> bool:
    return _load_hip() is not None


def hip_ops():
    """The compiled extension; raises on a GPU box when it is missing."""
    mod = _load_hip()
    if mod is None:
        raise RuntimeError(
            "dinov3_amd HIP extension (_hip_ops) is not built. On a GPU this "
            "framework refuses to fall back to eager PyTorch — run "
            "`python setup.py build_ext --inplace` (or __graft_entry__.build())."
        )
    return mod


def use_hip(x: torch.Tensor) -> bool:
    """True when `x` lives on a HIP device (and the extension must be used)."""
    if not x.is_cuda:
        return False
    if os.environ.get("DINOV3_DISABLE_HIP", "0") == "1":
        return False
    return True
