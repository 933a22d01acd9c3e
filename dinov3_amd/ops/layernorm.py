"""LayerNorm / RMSNorm.

GPU path: fused single-pass row kernels over D (fp32 accumulation, bf16 I/O),
csrc/norms.hip. CPU path / numerics oracle: torch.nn.functional in fp32.

Reference semantics: dinov3_jax/layers/rms_norm.py:17-28 and the
layernorm/layernormbf16 registry at vision_transformer.py:39-43.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from typing import Optional


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        from . import hip_ops

        ops = hip_ops()
        y, mean, rstd = ops.layernorm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        from . import hip_ops

        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = hip_ops().layernorm_bwd(dy.contiguous(), x, weight, mean, rstd)
        return dx, dw, db, None


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        from . import hip_ops

        y, rstd = hip_ops().rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        from . import hip_ops

        x, weight, rstd = ctx.saved_tensors
        dx, dw = hip_ops().rmsnorm_bwd(dy.contiguous(), x, weight, rstd)
        return dx, dw, None


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor], eps: float = 1e-6) -> torch.Tensor:
    from . import use_hip

    if use_hip(x):
        if bias is None:
            bias = torch.zeros_like(weight)
        return _LayerNormFn.apply(x.contiguous(), weight, bias, eps)
    # fp32 reference path
    out = F.layer_norm(x.float(), (x.shape[-1],), weight.float(), None if bias is None else bias.float(), eps)
    return out.to(x.dtype)


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    from . import use_hip

    if use_hip(x):
        return _RMSNormFn.apply(x.contiguous(), weight, eps)
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(dim=-1, keepdim=True) + eps) * weight.float()
    return y.to(x.dtype)
