"""Fused LayerScale + residual add: out = x + gamma * res."""

from __future__ import annotations

import torch


class _LsAxpyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, res, gamma):
        from . import hip_ops

        out = hip_ops().ls_axpy_fwd(x, res, gamma)
        ctx.save_for_backward(res, gamma)
        return out

    @staticmethod
    def backward(ctx, dout):
        from . import hip_ops

        res, gamma = ctx.saved_tensors
        dres, dgamma = hip_ops().ls_axpy_bwd(dout.contiguous(), res, gamma)
        return dout, dres, dgamma


def ls_axpy(x: torch.Tensor, res: torch.Tensor, gamma: torch.Tensor) -> torch.Tensor:
    from . import use_hip

    if use_hip(x) and x.shape[-1] % 8 == 0:
        return _LsAxpyFn.apply(x.contiguous(), res.contiguous(), gamma.contiguous())
    return x + gamma * res
