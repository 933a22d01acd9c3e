"""Patch embedding: p x p / stride-p conv == per-patch GEMM.

Weight stored as a Linear over the flattened patch [p*p*C, D] so the op lowers
to one [B*N, p*p*C] x [p*p*C, D] MFMA GEMM with a strided A-load instead of an
im2col conv. Input is NCHW (the collate layout); the patchify view is a
reshape/permute that the GPU path fuses into the GEMM's A read.

Reference: dinov3_jax/layers/patch_embed.py:21-56.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn


class PatchEmbed(nn.Module):
    def __init__(self, img_size: int = 224, patch_size: int = 16, in_chans: int = 3,
                 embed_dim: int = 768, norm_layer: Optional[nn.Module] = None, flatten_embedding: bool = True):
        super().__init__()
        self.img_size = img_size
        self.patch_size = patch_size
        self.in_chans = in_chans
        self.embed_dim = embed_dim
        self.flatten_embedding = flatten_embedding
        self.proj = nn.Linear(patch_size * patch_size * in_chans, embed_dim, bias=True)
        self.norm = norm_layer if norm_layer is not None else nn.Identity()

    def patchify(self, x: torch.Tensor) -> Tuple[torch.Tensor, int, int]:
        """[B, C, H, W] -> [B, H'/p * W'/p, C*p*p] rows in (c, dy, dx) order —
        the conv-weight-native flattening, so Meta conv weights map by a plain
        reshape and each 16-element k-slice is one contiguous pixel run."""
        B, C, H, W = x.shape
        p = self.patch_size
        assert H % p == 0 and W % p == 0, f"input {H}x{W} not divisible by patch {p}"
        hp, wp = H // p, W // p
        x = x.reshape(B, C, hp, p, wp, p)
        x = x.permute(0, 2, 4, 1, 3, 5)  # B, hp, wp, C, p, p
        return x.reshape(B, hp * wp, C * p * p), hp, wp

    def forward(self, x: torch.Tensor) -> Tuple[torch.Tensor, int, int]:
        from ..ops.patch_embed_op import patch_embed_gemm

        B, C, H, W = x.shape
        p = self.patch_size
        hp, wp = H // p, W // p
        out = patch_embed_gemm(x, self.proj.weight, self.proj.bias, p, self.patchify)
        out = self.norm(out)
        if not self.flatten_embedding:
            out = out.reshape(out.shape[0], hp, wp, self.embed_dim)
        return out, hp, wp
