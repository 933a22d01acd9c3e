"""Fused residual update: LayerScale gamma and the producing GEMM's bias
folded into the residual add / drop-path scatter (round-2 path, gated).

With DINOV3_FUSED_RESIDUAL=1 the transformer block computes the attention
proj and MLP fc2 GEMMs WITHOUT bias epilogue and folds the bias into the
residual kernel instead:

  dense:     out = flat + gamma * (res + bias)           (ls_axpy kernel)
  drop-path: flat[idx[r]] += scale[r] * gamma * (res[r] + bias)
                                                          (ls_scatter kernel)

This removes per-block on the bench path: the standalone LayerScale
multiply, the bias-grad `sum` reduction of proj/fc2 (their dbias rides the
residual kernel's existing column reduction), and one full-buffer clone
(the scatter mutates `flat` in place; the op is linear in flat so backward
is the identity — safe because no autograd node saves the flat buffer).

CPU fallback is plain torch with identical math, so the gated path is
equivalence-tested on CPU; GPU numerics in tests/test_ops_gpu.py (gated).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

__all__ = ["fused_residual_enabled", "ls_axpy_bias", "ls_scatter_add_rows"]


# Default ON since round 2: the round-1 regression was ls_scatter_bwd's
# inter-block atomic chains (profiles/r2_fused_residual_diagnosis.md); with
# the capped launchers the fused path wins on hardware (r2_gpu5: 449.4 vs
# 446.0 img/s before row batching). DINOV3_FUSED_RESIDUAL=0 disables.
_DEFAULT_ON = True


def fused_residual_enabled() -> bool:
    v = os.environ.get("DINOV3_FUSED_RESIDUAL")
    if v is None:
        return _DEFAULT_ON
    return v == "1"


class _LsAxpyBiasFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, res, gamma, bias):
        from . import hip_ops

        out = hip_ops().ls_axpy_bias_fwd(x, res, gamma, bias)
        ctx.save_for_backward(res, gamma, bias)
        return out

    @staticmethod
    def backward(ctx, dout):
        from . import hip_ops

        res, gamma, bias = ctx.saved_tensors
        dres, dgamma, dbias = hip_ops().ls_axpy_bias_bwd(dout.contiguous(), res, gamma, bias)
        return dout, dres, dgamma, dbias


def ls_axpy_bias(x: torch.Tensor, res: torch.Tensor, gamma: Optional[torch.Tensor],
                 bias: Optional[torch.Tensor]) -> torch.Tensor:
    """out = x + gamma * (res + bias); gamma/bias optional (1 / 0)."""
    from . import use_hip
    from .ls_axpy import ls_axpy

    if bias is None:
        if gamma is None:
            return x + res
        return ls_axpy(x, res, gamma)
    if use_hip(x) and x.shape[-1] % 8 == 0 and gamma is not None:
        return _LsAxpyBiasFn.apply(x.contiguous(), res.contiguous(),
                                   gamma.contiguous(), bias.contiguous())
    g = gamma if gamma is not None else 1.0
    return x + g * (res + bias)


class _LsScatterAddFn(torch.autograd.Function):
    """flat[idx] += scale * gamma * (res + bias), in place (linear in flat)."""

    @staticmethod
    def forward(ctx, flat, idx, res, gamma, bias, scale):
        from . import hip_ops

        hip_ops().ls_scatter_add_(flat, idx, res.contiguous(), gamma, bias, scale)
        ctx.save_for_backward(idx, res, gamma, bias, scale)
        ctx.mark_dirty(flat)
        return flat

    @staticmethod
    def backward(ctx, dout):
        from . import hip_ops

        idx, res, gamma, bias, scale = ctx.saved_tensors
        dres, dgamma, dbias = hip_ops().ls_scatter_bwd(
            dout.contiguous(), idx, res, gamma, bias, scale)
        return dout, None, dres, dgamma, dbias, None


def ls_scatter_add_rows(flat: torch.Tensor, idx: torch.Tensor, res: torch.Tensor,
                        gamma: Optional[torch.Tensor], bias: Optional[torch.Tensor],
                        scale: torch.Tensor) -> torch.Tensor:
    """flat.index_add(0, idx, scale[:, None] * gamma * (res + bias))."""
    from . import use_hip

    if (use_hip(flat) and flat.shape[-1] % 8 == 0 and gamma is not None
            and bias is not None):
        return _LsScatterAddFn.apply(flat, idx, res, gamma.contiguous(),
                                     bias.contiguous(), scale.float().contiguous())
    val = res if bias is None else res + bias
    if gamma is not None:
        val = gamma * val
    return flat.index_add(0, idx, (val * scale.unsqueeze(1)).to(flat.dtype))
