"""Multi-crop flat-buffer attention core (fused RoPE + FMHA on raw QKV).

The multi-crop forward keeps every token of every crop group in ONE flat
[R, D] buffer (global rows first, then local rows — the layout cat_keep_shapes
produces). The qkv GEMM runs once on the flat buffer; this Function then runs
the fused rope+attention kernel per crop group directly on contiguous slices
of the qkv output ([B, N, 3, H, hd] views) and writes each group's context
into a slice of one flat output — zero permutes, zero concats, one dqkv flat
buffer in backward.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

# group meta: (row_offset, B, N, sin, cos, prefix)
GroupMeta = Tuple[int, int, int, Optional[torch.Tensor], Optional[torch.Tensor], int]

_EMPTY = {}


def _empty(device):
    key = str(device)
    if key not in _EMPTY:
        _EMPTY[key] = torch.empty(0, device=device)
    return _EMPTY[key]


class _FlatMultiFmha(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv_flat: torch.Tensor, num_heads: int, metas: List[GroupMeta]):
        from . import hip_ops

        ops = hip_ops()
        R, threeD = qkv_flat.shape
        D = threeD // 3
        hd = D // num_heads
        out_flat = torch.empty(R, D, dtype=qkv_flat.dtype, device=qkv_flat.device)
        lses = []
        e = _empty(qkv_flat.device)
        for (off, B, N, sin, cos, prefix) in metas:
            qkv_g = qkv_flat[off: off + B * N].view(B, N, 3, num_heads, hd)
            o_view = out_flat[off: off + B * N].view(B, N, num_heads, hd)
            _, lse = ops.fmha_rope_fwd_out(qkv_g, sin if sin is not None else e,
                                           cos if cos is not None else e, prefix, o_view)
            lses.append(lse)
        ctx.save_for_backward(qkv_flat, out_flat, *lses)
        ctx.metas = metas
        ctx.num_heads = num_heads
        return out_flat

    @staticmethod
    def backward(ctx, dout_flat):
        from . import hip_ops

        ops = hip_ops()
        qkv_flat, out_flat, *lses = ctx.saved_tensors
        num_heads = ctx.num_heads
        R, threeD = qkv_flat.shape
        D = threeD // 3
        hd = D // num_heads
        dout_flat = dout_flat.contiguous()
        dqkv_flat = torch.empty_like(qkv_flat)
        e = _empty(qkv_flat.device)
        for (off, B, N, sin, cos, prefix), lse in zip(ctx.metas, lses):
            qkv_g = qkv_flat[off: off + B * N].view(B, N, 3, num_heads, hd)
            o_g = out_flat[off: off + B * N].view(B, N, num_heads, hd)
            do_g = dout_flat[off: off + B * N].view(B, N, num_heads, hd)
            dqkv_g = dqkv_flat[off: off + B * N].view(B, N, 3, num_heads, hd)
            ops.fmha_rope_bwd_out(do_g, qkv_g, o_g, lse,
                                  sin if sin is not None else e,
                                  cos if cos is not None else e, prefix, dqkv_g)
        return dqkv_flat, None, None


def _flat_fmha_ref(qkv_flat: torch.Tensor, num_heads: int, metas: List[GroupMeta]) -> torch.Tensor:
    """CPU / reference path: same semantics with plain torch ops."""
    from .fmha import fmha_ref
    from .rope import rope_apply

    R, threeD = qkv_flat.shape
    D = threeD // 3
    hd = D // num_heads
    outs = []
    for (off, B, N, sin, cos, prefix) in metas:
        qkv_g = qkv_flat[off: off + B * N].view(B, N, 3, num_heads, hd)
        q, k, v = qkv_g.permute(2, 0, 3, 1, 4).unbind(0)  # [B, H, N, hd]
        if sin is not None:
            q = rope_apply(q, sin, cos, prefix)
            k = rope_apply(k, sin, cos, prefix)
        o = fmha_ref(q, k, v)  # [B, H, N, hd]
        outs.append(o.permute(0, 2, 1, 3).reshape(B * N, D))
    return torch.cat(outs, dim=0)


def flat_multi_fmha(qkv_flat: torch.Tensor, num_heads: int, metas: List[GroupMeta]) -> torch.Tensor:
    from . import use_hip

    if use_hip(qkv_flat) and qkv_flat.dtype == torch.bfloat16:
        return _FlatMultiFmha.apply(qkv_flat, num_heads, metas)
    return _flat_fmha_ref(qkv_flat, num_heads, metas)
