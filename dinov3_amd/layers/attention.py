"""Self-attention: fused qkv GEMM -> RoPE apply -> FMHA -> out projection.

Reference: dinov3_jax/layers/attention.py:49-133. mask_k_bias (vit7b configs)
means the k slice of the qkv bias is held at zero (the reference expresses
this with a NaN-filled buffer times bias; SURVEY K4 intended semantics).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import fmha, rope_apply
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

    def _attend(self, qkv: torch.Tensor, rope: Optional[RopeSinCos]) -> torch.Tensor:
        """qkv: [B, N, 3*dim] -> context [B, N, dim]."""
        B, N, _ = qkv.shape
        qkv = qkv.reshape(B, N, 3, self.num_heads, self.head_dim)
        q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)  # each [B, H, N, hd]
        if rope is not None:
            sin, cos = rope
            prefix = N - sin.shape[0]
            assert prefix >= 0
            q = rope_apply(q, sin, cos, prefix)
            k = rope_apply(k, sin, cos, prefix)
        x = fmha(q.contiguous(), k.contiguous(), v.contiguous())  # [B, H, N, hd]
        return x.permute(0, 2, 1, 3).reshape(B, N, self.dim)

    def forward(self, x: torch.Tensor, rope: Optional[RopeSinCos] = None) -> torch.Tensor:
        qkv = self.qkv(x)
        ctx = self._attend(qkv, rope)
        return self.proj_drop(self.proj(ctx))

    def forward_list(self, x_list: List[torch.Tensor],
                     rope_list: Optional[List[Optional[RopeSinCos]]] = None) -> List[torch.Tensor]:
        """Multi-crop forward: qkv + out-proj GEMMs run on the concatenated
        token batch; FMHA runs per crop group (different seqlens)."""
        if rope_list is None:
            rope_list = [None] * len(x_list)
        flat, shapes, counts = cat_keep_shapes(x_list)
        qkv_flat = self.qkv(flat)
        qkv_list = uncat_with_shapes(qkv_flat, shapes, counts)
        ctx_list = [self._attend(qkv, rope) for qkv, rope in zip(qkv_list, rope_list)]
        ctx_flat, shapes2, counts2 = cat_keep_shapes(ctx_list)
        out_flat = self.proj_drop(self.proj(ctx_flat))
        return uncat_with_shapes(out_flat, shapes2, counts2)
