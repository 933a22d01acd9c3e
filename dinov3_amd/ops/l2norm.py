"""Row-wise L2 normalization (DINO head bottleneck, KoLeo, Gram — SURVEY K15/K20/K21)."""

from __future__ import annotations

import torch


class _L2NormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, eps):
        from . import hip_ops

        y, rnorm = hip_ops().l2norm_fwd(x, eps)
        ctx.save_for_backward(y, rnorm)
        ctx.eps = eps
        return y

    @staticmethod
    def backward(ctx, dy):
        from . import hip_ops

        y, rnorm = ctx.saved_tensors
        dx = hip_ops().l2norm_bwd(dy.contiguous(), y, rnorm, ctx.eps)
        return dx, None


def l2_normalize(x: torch.Tensor, eps: float = 1e-12, dim: int = -1) -> torch.Tensor:
    from . import use_hip

    if dim not in (-1, x.ndim - 1):
        return torch.nn.functional.normalize(x, dim=dim, eps=eps)
    if use_hip(x):
        return _L2NormFn.apply(x.contiguous(), eps)
    # semantics: x / (||x|| + eps)  (matches the reference DINO head,
    # dinov3_jax/layers/dino_head.py:79-81)
    xf = x.float()
    y = xf / (xf.norm(dim=-1, keepdim=True) + eps)
    return y.to(x.dtype)
