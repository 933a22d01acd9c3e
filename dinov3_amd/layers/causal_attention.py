"""Causal self-attention branch (parity with the reference's vestigial text
tower: dinov3_jax/layers/attention.py:135+, block.py CausalSelfAttentionBlock)."""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .ffn_layers import Mlp
from .norms import LayerNorm, LayerScale


class CausalSelfAttention(nn.Module):
    def __init__(self, dim: int, num_heads: int = 8, attn_drop: float = 0.0,
                 proj_bias: bool = True, qkv_bias: bool = True):
        super().__init__()
        assert dim % num_heads == 0
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.qkv = nn.Linear(dim, 3 * dim, bias=qkv_bias)
        self.proj = nn.Linear(dim, dim, bias=proj_bias)
        self.attn_drop = attn_drop

    def forward(self, x: torch.Tensor, is_causal: bool = True) -> torch.Tensor:
        B, N, D = x.shape
        qkv = self.qkv(x).reshape(B, N, 3, self.num_heads, self.head_dim)
        q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)
        s = torch.einsum("bhqd,bhkd->bhqk", q.float(), k.float()) / math.sqrt(self.head_dim)
        if is_causal:
            mask = torch.triu(torch.ones(N, N, device=x.device, dtype=torch.bool), diagonal=1)
            s = s.masked_fill(mask, float("-inf"))
        p = torch.softmax(s, dim=-1)
        if self.training and self.attn_drop > 0:
            p = F.dropout(p, p=self.attn_drop)
        o = torch.einsum("bhqk,bhkd->bhqd", p, v.float()).to(x.dtype)
        o = o.permute(0, 2, 1, 3).reshape(B, N, D)
        return self.proj(o)


class CausalSelfAttentionBlock(nn.Module):
    def __init__(self, dim: int, num_heads: int, ffn_ratio: float = 4.0,
                 ls_init_value: Optional[float] = None, is_causal: bool = True,
                 dropout_prob: float = 0.0):
        super().__init__()
        self.is_causal = is_causal
        self.attention_norm = LayerNorm(dim, eps=1e-6)
        self.attention = CausalSelfAttention(dim, num_heads=num_heads, attn_drop=dropout_prob)
        self.ls1 = LayerScale(dim, ls_init_value) if ls_init_value else nn.Identity()
        self.ffn_norm = LayerNorm(dim, eps=1e-6)
        self.feed_forward = Mlp(in_features=dim, hidden_features=int(dim * ffn_ratio))
        self.ls2 = LayerScale(dim, ls_init_value) if ls_init_value else nn.Identity()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x + self.ls1(self.attention(self.attention_norm(x), self.is_causal))
        return x + self.ls2(self.feed_forward(self.ffn_norm(x)))
