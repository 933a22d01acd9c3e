"""Norm layers + LayerScale over the fused HIP ops.

Reference: dinov3_jax/layers/rms_norm.py, layer_scale.py, and the norm
registry at vision_transformer.py:39-43 (layernorm eps 1e-6, layernormbf16
eps 1e-5, rmsnorm).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import layer_norm, rms_norm


class LayerNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6, **kwargs):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return layer_norm(x, self.weight, self.bias, self.eps)


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6, **kwargs):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return rms_norm(x, self.weight, self.eps)


class LayerScale(nn.Module):
    def __init__(self, dim: int, init_values: float = 1e-5):
        super().__init__()
        self.gamma = nn.Parameter(init_values * torch.ones(dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x * self.gamma


NORM_LAYERS = {
    "layernorm": lambda dim: LayerNorm(dim, eps=1e-6),
    "layernormbf16": lambda dim: LayerNorm(dim, eps=1e-5),
    "rmsnorm": lambda dim: RMSNorm(dim, eps=1e-6),
}
