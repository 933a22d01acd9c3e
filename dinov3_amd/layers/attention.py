"""Self-attention on the flat multi-crop token buffer.

qkv GEMM and out-projection run once on the concatenated [R, D] buffer of all
crop groups; the fused rope+FMHA kernel consumes contiguous per-group slices
of the qkv output directly (ops/flat_attention.py) — no permutes or concats
on the hot path.

Reference semantics: dinov3_jax/layers/attention.py:49-133. mask_k_bias
(vit7b configs) keeps the k third of the qkv bias at zero.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.flat_attention import GroupMeta, flat_multi_fmha
from ..utils.utils import cat_keep_shapes, uncat_with_shapes

RopeSinCos = Tuple[torch.Tensor, torch.Tensor]


class LinearKMaskedBias(nn.Linear):
    """qkv linear whose k-bias third is masked to zero (non-trainable zeros)."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__(in_features, out_features, bias=bias)
        assert out_features % 3 == 0
        mask = torch.ones(out_features)
        third = out_features // 3
        mask[third: 2 * third] = 0.0
        self.register_buffer("bias_mask", mask, persistent=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b = self.bias * self.bias_mask.to(self.bias.dtype) if self.bias is not None else None
        return F.linear(x, self.weight, b)


class SelfAttention(nn.Module):
    def __init__(
        self,
        dim: int,
        num_heads: int = 8,
        qkv_bias: bool = False,
        proj_bias: bool = True,
        attn_drop: float = 0.0,
        proj_drop: float = 0.0,
        mask_k_bias: bool = False,
    ):
        super().__init__()
        assert dim % num_heads == 0
        self.dim = dim
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        linear_class = LinearKMaskedBias if mask_k_bias else nn.Linear
        self.qkv = linear_class(dim, dim * 3, bias=qkv_bias)
        self.proj = nn.Linear(dim, dim, bias=proj_bias)
        self.proj_drop = nn.Dropout(proj_drop) if proj_drop > 0 else nn.Identity()

    def forward_flat(self, flat: torch.Tensor, metas: List[GroupMeta],
                     skip_proj_bias: bool = False) -> torch.Tensor:
        """flat: [R, D] concatenated tokens of all crop groups.

        skip_proj_bias: leave the out-projection bias off the GEMM — the
        caller folds it into the residual kernel (ops/fused_residual.py)."""
        qkv_flat = self.qkv(flat)
        ctx_flat = flat_multi_fmha(qkv_flat, self.num_heads, metas)
        if skip_proj_bias:
            import torch.nn.functional as F

            return self.proj_drop(F.linear(ctx_flat, self.proj.weight))
        return self.proj_drop(self.proj(ctx_flat))

    @staticmethod
    def _meta_for(x: torch.Tensor, rope: Optional[RopeSinCos], offset: int) -> GroupMeta:
        B, N, _ = x.shape
        if rope is None:
            return (offset, B, N, None, None, 0)
        sin, cos = rope
        prefix = N - sin.shape[0]
        assert prefix >= 0
        return (offset, B, N, sin.float().contiguous(), cos.float().contiguous(), prefix)

    def forward(self, x: torch.Tensor, rope: Optional[RopeSinCos] = None) -> torch.Tensor:
        B, N, D = x.shape
        meta = self._meta_for(x, rope, 0)
        out = self.forward_flat(x.reshape(B * N, D), [meta])
        return out.reshape(B, N, D)

    def forward_list(self, x_list: List[torch.Tensor],
                     rope_list: Optional[List[Optional[RopeSinCos]]] = None) -> List[torch.Tensor]:
        if rope_list is None:
            rope_list = [None] * len(x_list)
        flat, shapes, counts = cat_keep_shapes(x_list)
        metas = []
        off = 0
        for x, rope in zip(x_list, rope_list):
            metas.append(self._meta_for(x, rope, off))
            off += x.shape[0] * x.shape[1]
        out_flat = self.forward_flat(flat, metas)
        return uncat_with_shapes(out_flat, shapes, counts)
