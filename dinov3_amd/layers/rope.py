"""Axial 2-D RoPE table.

Reference: dinov3_jax/layers/rope_position_encoding.py:17-123. The table is a
tiny [H*W, D_head] sin/cos pair computed per forward in fp32 (train-time
coordinate shift/jitter/rescale augments use the host RNG); the hot work — the
rotate-half application to q/k — lives in ops/rope.py.

Note the reference's "min" normalization uses max(H, W) (bug);
we implement min(H, W) as intended.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn


class RopePositionEmbedding(nn.Module):
    def __init__(
        self,
        embed_dim: int,
        num_heads: int,
        base: Optional[float] = 100.0,
        min_period: Optional[float] = None,
        max_period: Optional[float] = None,
        normalize_coords: str = "separate",
        shift_coords: Optional[float] = None,
        jitter_coords: Optional[float] = None,
        rescale_coords: Optional[float] = None,
        dtype: torch.dtype = torch.float32,
    ):
        super().__init__()
        assert embed_dim % (4 * num_heads) == 0
        both = min_period is not None and max_period is not None
        if (base is None) == (not both):
            raise ValueError("Either `base` or `min_period`+`max_period` must be provided.")
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.normalize_coords = normalize_coords
        self.shift_coords = shift_coords
        self.jitter_coords = jitter_coords
        self.rescale_coords = rescale_coords
        self.dtype = dtype
        d_head = embed_dim // num_heads
        if base is not None:
            periods = base ** (2.0 * torch.arange(d_head // 4, dtype=torch.float32) / (d_head / 2.0))
        else:
            ratio = max_period / min_period
            exponents = torch.linspace(0.0, 1.0, d_head // 4)
            periods = (ratio**exponents) / ratio * max_period
        self.register_buffer("periods", periods, persistent=False)

    def forward(self, H: int, W: int, training: bool = False,
                device: Optional[torch.device] = None) -> Tuple[torch.Tensor, torch.Tensor]:
        dev = device if device is not None else self.periods.device
        dd = {"device": dev, "dtype": torch.float32}
        if self.normalize_coords == "max":
            denom_h = denom_w = float(max(H, W))
        elif self.normalize_coords == "min":
            denom_h = denom_w = float(min(H, W))
        elif self.normalize_coords == "separate":
            denom_h, denom_w = float(H), float(W)
        else:
            raise ValueError(f"Unknown normalize_coords: {self.normalize_coords}")
        coords_h = torch.arange(0.5, H, **dd) / denom_h
        coords_w = torch.arange(0.5, W, **dd) / denom_w
        coords = torch.stack(torch.meshgrid(coords_h, coords_w, indexing="ij"), dim=-1).reshape(-1, 2)
        coords = 2.0 * coords - 1.0

        if training:
            if self.shift_coords is not None:
                shift = torch.empty(2, **dd).uniform_(-self.shift_coords, self.shift_coords)
                coords = coords + shift[None, :]
            if self.jitter_coords is not None:
                jmax = math.log(self.jitter_coords)
                jitter = torch.exp(torch.empty(2, **dd).uniform_(-jmax, jmax))
                coords = coords * jitter[None, :]
            if self.rescale_coords is not None:
                rmax = math.log(self.rescale_coords)
                rescale = torch.exp(torch.empty(1, **dd).uniform_(-rmax, rmax))
                coords = coords * rescale

        periods = self.periods.to(dev, torch.float32)
        angles = 2.0 * math.pi * coords[:, :, None] / periods[None, None, :]  # [HW, 2, D/4]
        angles = angles.reshape(angles.shape[0], -1)
        angles = torch.cat([angles, angles], dim=-1)  # [HW, D_head]
        return torch.sin(angles), torch.cos(angles)
