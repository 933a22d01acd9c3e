"""DINO / iBOT projection head.

MLP (1 or 3 layers) -> L2 bottleneck normalize -> bias-free prototype layer.
Reference: dinov3_jax/layers/dino_head.py:15-85.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import l2_normalize


def _trunc_normal_init(module: nn.Linear) -> None:
    nn.init.trunc_normal_(module.weight, std=0.02, a=-1.0, b=1.0)
    if module.bias is not None:
        nn.init.zeros_(module.bias)


class DINOHead(nn.Module):
    def __init__(self, in_dim: int, out_dim: int, use_bn: bool = False, nlayers: int = 3,
                 hidden_dim: int = 2048, bottleneck_dim: int = 256, mlp_bias: bool = True):
        super().__init__()
        nlayers = max(nlayers, 1)
        if nlayers == 1:
            layers = [nn.Linear(in_dim, bottleneck_dim, bias=mlp_bias)]
        else:
            layers = [nn.Linear(in_dim, hidden_dim, bias=mlp_bias), nn.GELU(approximate="tanh")]
            for _ in range(nlayers - 2):
                layers += [nn.Linear(hidden_dim, hidden_dim, bias=mlp_bias), nn.GELU(approximate="tanh")]
            layers.append(nn.Linear(hidden_dim, bottleneck_dim, bias=mlp_bias))
        self.mlp = nn.Sequential(*layers)
        self.last_layer = nn.Linear(bottleneck_dim, out_dim, bias=False)
        for m in self.modules():
            if isinstance(m, nn.Linear):
                _trunc_normal_init(m)

    def forward(self, x: torch.Tensor, no_last_layer: bool = False, only_last_layer: bool = False) -> torch.Tensor:
        if not only_last_layer:
            x = self.mlp(x)
            eps = 1e-6 if x.dtype == torch.float16 else 1e-12
            x = l2_normalize(x, eps=eps)
        if not no_last_layer:
            x = self.last_layer(x)
        return x
