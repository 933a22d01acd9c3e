"""Fused bias+GELU (MLP epilogue) and SwiGLU gate (K9).

The reference Mlp applies an EXTRA activation after fc2 (a bug,
dinov3_jax/layers/ffn_layers.py:47-48) — we implement standard
ViT MLP semantics: fc1 -> gelu -> drop -> fc2 -> drop.
"""

from __future__ import annotations

import torch


class _BiasGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        from . import hip_ops

        y = hip_ops().bias_gelu_fwd(x, bias)
        ctx.save_for_backward(x, bias)
        return y

    @staticmethod
    def backward(ctx, dy):
        from . import hip_ops

        x, bias = ctx.saved_tensors
        dx, dbias = hip_ops().bias_gelu_bwd(dy.contiguous(), x, bias)
        return dx, dbias


class _SwigluGateFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x12):
        from . import hip_ops

        y = hip_ops().swiglu_fwd(x12)
        ctx.save_for_backward(x12)
        return y

    @staticmethod
    def backward(ctx, dy):
        from . import hip_ops

        (x12,) = ctx.saved_tensors
        return hip_ops().swiglu_bwd(dy.contiguous(), x12)


def _gelu_tanh_ref(x: torch.Tensor) -> torch.Tensor:
    return torch.nn.functional.gelu(x, approximate="tanh")


def bias_gelu(x: torch.Tensor, bias: torch.Tensor | None) -> torch.Tensor:
    """y = gelu_tanh(x + bias), fused on GPU."""
    from . import use_hip

    if use_hip(x):
        if bias is None:
            bias = torch.zeros(x.shape[-1], dtype=x.dtype, device=x.device)
        return _BiasGeluFn.apply(x.contiguous(), bias.contiguous())
    xf = x.float() + (0.0 if bias is None else bias.float())
    return _gelu_tanh_ref(xf).to(x.dtype)


def swiglu_gate(x12: torch.Tensor) -> torch.Tensor:
    """x12 = [.., 2H] (x1|x2) -> silu(x1) * x2, fused on GPU."""
    from . import use_hip

    if use_hip(x12):
        return _SwigluGateFn.apply(x12.contiguous())
    h = x12.shape[-1] // 2
    x1, x2 = x12[..., :h].float(), x12[..., h:].float()
    return (torch.nn.functional.silu(x1) * x2).to(x12.dtype)
