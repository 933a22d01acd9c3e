"""Pre-norm transformer block with LayerScale and stochastic depth as
batch-subset compute.

Reference: dinov3_jax/layers/block.py:22-208. Stochastic depth runs
attn/ffn only on a random ~b*(1-p) row subset and scatter-adds the rescaled
residual — a throughput feature we keep. Norms and FFN run on the
concatenated multi-crop token batch (cat_keep_shapes), attention per crop
group.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from ..utils.utils import cat_keep_shapes, uncat_with_shapes
from .attention import RopeSinCos, SelfAttention
from .ffn_layers import Mlp
from .norms import LayerScale


def _subset_indices(b: int, keep: int, device) -> torch.Tensor:
    return torch.randperm(b, device=device)[:keep]


class SelfAttentionBlock(nn.Module):
    def __init__(
        self,
        dim: int,
        num_heads: int,
        ffn_ratio: float = 4.0,
        qkv_bias: bool = False,
        proj_bias: bool = True,
        ffn_bias: bool = True,
        drop: float = 0.0,
        attn_drop: float = 0.0,
        init_values: Optional[float] = None,
        drop_path: float = 0.0,
        norm_layer=None,
        ffn_layer=None,
        mask_k_bias: bool = False,
    ):
        super().__init__()
        from .norms import LayerNorm

        norm_layer = norm_layer or (lambda d: LayerNorm(d, eps=1e-6))
        ffn_layer = ffn_layer or Mlp
        self.norm1 = norm_layer(dim)
        self.attn = SelfAttention(
            dim, num_heads=num_heads, qkv_bias=qkv_bias, proj_bias=proj_bias,
            attn_drop=attn_drop, proj_drop=drop, mask_k_bias=mask_k_bias,
        )
        self.ls1 = LayerScale(dim, init_values) if init_values is not None else nn.Identity()
        self.norm2 = norm_layer(dim)
        self.mlp = ffn_layer(in_features=dim, hidden_features=int(dim * ffn_ratio), drop=drop, use_bias=ffn_bias)
        self.ls2 = LayerScale(dim, init_values) if init_values is not None else nn.Identity()
        self.sample_drop_ratio = drop_path

    @staticmethod
    def _index_rope(rope: Optional[RopeSinCos], indices: torch.Tensor) -> Optional[RopeSinCos]:
        if rope is None:
            return None
        sin, cos = rope
        if sin.ndim == 4:  # per-sample tables (coord augments per sample) — not used currently
            return sin[indices], cos[indices]
        return rope  # shared [P, hd] table: subset along batch leaves it unchanged

    def forward(self, x: torch.Tensor, rope: Optional[RopeSinCos] = None) -> torch.Tensor:
        return self.forward_list([x], [rope])[0]

    def forward_list(self, x_list: List[torch.Tensor],
                     rope_list: Optional[List[Optional[RopeSinCos]]] = None) -> List[torch.Tensor]:
        if rope_list is None:
            rope_list = [None] * len(x_list)
        use_droppath = self.training and self.sample_drop_ratio > 0.0
        if not use_droppath:
            flat, shapes, counts = cat_keep_shapes(x_list)
            n1 = uncat_with_shapes(self.norm1(flat), shapes, counts)
            attn_res = self.attn.forward_list(n1, rope_list)
            x_list = [x + self.ls1(r) for x, r in zip(x_list, attn_res)]
            flat, shapes, counts = cat_keep_shapes(x_list)
            mlp_res = self.mlp(self.norm2(flat))
            mlp_res = self.ls2(mlp_res)
            return [x + r for x, r in zip(x_list, uncat_with_shapes(mlp_res, shapes, counts))]

        # stochastic depth: subset-compute with scatter-add rescaled residual
        b_list = [x.shape[0] for x in x_list]
        keeps = [max(int(b * (1.0 - self.sample_drop_ratio)), 1) for b in b_list]
        scales = [b / k for b, k in zip(b_list, keeps)]

        idx1 = [_subset_indices(b, k, x.device) for b, k, x in zip(b_list, keeps, x_list)]
        sub1 = [x[i] for x, i in zip(x_list, idx1)]
        rope_sub = [self._index_rope(r, i) for r, i in zip(rope_list, idx1)]
        flat, shapes, counts = cat_keep_shapes(sub1)
        n1 = uncat_with_shapes(self.norm1(flat), shapes, counts)
        attn_res = self.attn.forward_list(n1, rope_sub)
        x_list = [
            x.index_add(0, i, (self.ls1(r) * s).to(x.dtype))
            for x, i, r, s in zip(x_list, idx1, attn_res, scales)
        ]

        idx2 = [_subset_indices(b, k, x.device) for b, k, x in zip(b_list, keeps, x_list)]
        sub2 = [x[i] for x, i in zip(x_list, idx2)]
        flat, shapes, counts = cat_keep_shapes(sub2)
        mlp_res = self.ls2(self.mlp(self.norm2(flat)))
        mlp_list = uncat_with_shapes(mlp_res, shapes, counts)
        return [
            x.index_add(0, i, (r * s).to(x.dtype))
            for x, i, r, s in zip(x_list, idx2, mlp_list, scales)
        ]
