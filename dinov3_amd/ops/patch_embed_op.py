"""Patch-embed GEMM op: hand-written MFMA kernel with the NCHW
patchify gather fused into the A-operand staging (no permute copy).

The input images carry no gradient, so backward only needs the weight/bias
grads: the patch rows are recomputed with the cheap patchify view and the
wgrad runs as one hipBLASLt GEMM.
"""

from __future__ import annotations

import torch


class _PatchEmbedFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, patch, patchify):
        from . import hip_ops

        out = hip_ops().patch_embed_fwd(x, weight, bias, patch)
        ctx.save_for_backward(x)
        ctx.patchify = patchify
        return out

    @staticmethod
    def backward(ctx, dout):
        (x,) = ctx.saved_tensors
        rows, _, _ = ctx.patchify(x)  # [B, N, K]
        rows = rows.reshape(-1, rows.shape[-1])
        g = dout.reshape(-1, dout.shape[-1])
        dw = g.transpose(0, 1) @ rows
        db = g.sum(dim=0)
        return None, dw, db, None, None


def patch_embed_gemm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
                     patch: int, patchify) -> torch.Tensor:
    """x [B,C,H,W] -> [B, N, D]."""
    from . import use_hip

    # any patch size with C=3: P=16 takes the vector-staged kernel, others
    # (14 for ViT-g/14, 8, ...) the gather-staged generic variant
    if (use_hip(x) and x.dtype == torch.bfloat16 and x.shape[1] == 3
            and not x.requires_grad):
        return _PatchEmbedFn.apply(x.contiguous(), weight.contiguous(),
                                   bias.contiguous(), patch, patchify)
    rows, _, _ = patchify(x)
    return torch.nn.functional.linear(rows.to(weight.dtype), weight, bias)
