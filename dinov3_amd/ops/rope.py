"""Fused RoPE application on q/k with prefix-token exclusion.

Semantics (reference dinov3_jax/layers/attention.py:14-20,69-90): rotate-half
RoPE — out = x*cos + [-x2, x1]*sin where x1/x2 are the two halves of the head
dim — applied in fp32 to all tokens AFTER the first `prefix` (cls + storage)
tokens, then cast back to the input dtype. sin/cos are [P, hd] tables for the
P patch tokens.

GPU path: one fused kernel over q and k (csrc/rope.hip) — reads qkv once,
writes rotated q/k; no dtype round-trip tensors materialized.
"""

from __future__ import annotations

from typing import Tuple

import torch


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, sin, cos, prefix):
        from . import hip_ops

        y = hip_ops().rope_fwd(x, sin, cos, prefix)
        ctx.save_for_backward(sin, cos)
        ctx.prefix = prefix
        return y

    @staticmethod
    def backward(ctx, dy):
        from . import hip_ops

        sin, cos = ctx.saved_tensors
        # d/dx (x*cos + rot(x)*sin) => dx = dy*cos + rot^{-1}(dy)*sin, where
        # rot^{-1}([a,b]) = [b,-a] = rot with negated sin.
        dx = hip_ops().rope_fwd(dy.contiguous(), (-sin).contiguous(), cos, ctx.prefix)
        return dx, None, None, None


def _rotate_half(x: torch.Tensor) -> torch.Tensor:
    h = x.shape[-1] // 2
    return torch.cat([-x[..., h:], x[..., :h]], dim=-1)


def rope_apply(x: torch.Tensor, sin: torch.Tensor, cos: torch.Tensor, prefix: int) -> torch.Tensor:
    """x: [B, H, N, hd]; sin/cos: [N - prefix, hd]. Returns rotated x."""
    from . import use_hip

    if use_hip(x):
        return _RopeFn.apply(x.contiguous(), sin.contiguous(), cos.contiguous(), prefix)
    xf = x.float()
    head = xf[..., :prefix, :]
    tail = xf[..., prefix:, :]
    rot = tail * cos.float() + _rotate_half(tail) * sin.float()
    return torch.cat([head, rot], dim=-2).to(x.dtype)


def rope_apply_qk(q: torch.Tensor, k: torch.Tensor, sin: torch.Tensor, cos: torch.Tensor,
                  prefix: int) -> Tuple[torch.Tensor, torch.Tensor]:
    return rope_apply(q, sin, cos, prefix), rope_apply(k, sin, cos, prefix)
