"""ConvNeXt backbone with the DINO output-dict contract.

The reference ships a broken ConvNeXt (raises at convnext.py:83, syntax errors
at :227 — SURVEY §8 I1); this is a working implementation exposing the same
capability surface: forward_features returns the x_norm_clstoken /
x_norm_patchtokens dict the SSL meta-arch consumes, sizes tiny..large.
"""

from __future__ import annotations

from typing import Optional, Sequence

import torch
import torch.nn as nn

from ..ops import layer_norm


class _LayerNorm2d(nn.Module):
    """channels-first LayerNorm over C for [B, C, H, W]."""

    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = x.permute(0, 2, 3, 1)
        y = layer_norm(y, self.weight, self.bias, self.eps)
        return y.permute(0, 3, 1, 2)


class ConvNeXtBlock(nn.Module):
    def __init__(self, dim: int, drop_path: float = 0.0, layer_scale_init: float = 1e-6):
        super().__init__()
        self.dwconv = nn.Conv2d(dim, dim, kernel_size=7, padding=3, groups=dim)
        self.norm = nn.LayerNorm(dim, eps=1e-6)
        self.pwconv1 = nn.Linear(dim, 4 * dim)
        self.act = nn.GELU(approximate="tanh")
        self.pwconv2 = nn.Linear(4 * dim, dim)
        self.gamma = nn.Parameter(layer_scale_init * torch.ones(dim)) if layer_scale_init > 0 else None
        self.drop_path = drop_path

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shortcut = x
        y = self.dwconv(x)
        y = y.permute(0, 2, 3, 1)
        y = self.norm(y)
        y = self.pwconv2(self.act(self.pwconv1(y)))
        if self.gamma is not None:
            y = self.gamma * y
        y = y.permute(0, 3, 1, 2)
        if self.training and self.drop_path > 0:
            keep = 1.0 - self.drop_path
            mask = torch.rand(x.shape[0], 1, 1, 1, device=x.device) < keep
            y = y * mask.to(y.dtype) / keep
        return shortcut + y


class ConvNeXt(nn.Module):
    def __init__(
        self,
        in_chans: int = 3,
        depths: Sequence[int] = (3, 3, 9, 3),
        dims: Sequence[int] = (96, 192, 384, 768),
        drop_path_rate: float = 0.0,
        layer_scale_init: float = 1e-6,
        **kwargs,
    ):
        super().__init__()
        self.depths = list(depths)
        self.dims = list(dims)
        self.embed_dim = dims[-1]
        self.num_features = dims[-1]
        self.patch_size = 32  # total downsample factor
        self.n_storage_tokens = 0

        self.downsample_layers = nn.ModuleList()
        stem = nn.Sequential(nn.Conv2d(in_chans, dims[0], kernel_size=4, stride=4),
                             _LayerNorm2d(dims[0]))
        self.downsample_layers.append(stem)
        for i in range(3):
            self.downsample_layers.append(
                nn.Sequential(_LayerNorm2d(dims[i]),
                              nn.Conv2d(dims[i], dims[i + 1], kernel_size=2, stride=2))
            )

        rates = torch.linspace(0, drop_path_rate, sum(depths)).tolist()
        self.stages = nn.ModuleList()
        cur = 0
        for i in range(4):
            self.stages.append(nn.Sequential(*[
                ConvNeXtBlock(dims[i], drop_path=rates[cur + j], layer_scale_init=layer_scale_init)
                for j in range(depths[i])
            ]))
            cur += depths[i]
        self.norm = nn.LayerNorm(dims[-1], eps=1e-6)

    def forward_features(self, x: torch.Tensor, masks: Optional[torch.Tensor] = None):
        for i in range(4):
            x = self.downsample_layers[i](x)
            x = self.stages[i](x)
        B, C, H, W = x.shape
        tokens = x.permute(0, 2, 3, 1).reshape(B, H * W, C)
        tokens = self.norm(tokens)
        cls = tokens.mean(dim=1)
        return {
            "x_norm_clstoken": cls,
            "x_storage_tokens": tokens.new_zeros(B, 0, C),
            "x_norm_patchtokens": tokens,
            "x_prenorm": tokens,
            "masks": masks,
        }

    def forward(self, x, masks=None, is_training: bool = False):
        if isinstance(x, (list, tuple)):
            outs = [self.forward_features(t, m) for t, m in
                    zip(x, masks if masks is not None else [None] * len(x))]
            if is_training:
                return outs
            return [o["x_norm_clstoken"] for o in outs]
        out = self.forward_features(x, masks)
        if is_training:
            return out
        return out["x_norm_clstoken"]


def convnext_tiny(**kwargs):
    return ConvNeXt(depths=(3, 3, 9, 3), dims=(96, 192, 384, 768), **kwargs)


def convnext_small(**kwargs):
    return ConvNeXt(depths=(3, 3, 27, 3), dims=(96, 192, 384, 768), **kwargs)


def convnext_base(**kwargs):
    return ConvNeXt(depths=(3, 3, 27, 3), dims=(128, 256, 512, 1024), **kwargs)


def convnext_large(**kwargs):
    return ConvNeXt(depths=(3, 3, 27, 3), dims=(192, 384, 768, 1536), **kwargs)
