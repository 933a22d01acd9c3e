"""Small shared utilities (parity: dinov3_jax/utils/utils.py)."""

from __future__ import annotations

import random
from typing import List, Sequence, Tuple

import numpy as np
import torch


def cat_keep_shapes(tensors: Sequence[torch.Tensor]) -> Tuple[torch.Tensor, List[torch.Size], List[int]]:
    """Flatten each [B_i, N_i, D] tensor to rows and concatenate.

    Returns (concat [sum(B_i*N_i), D], original shapes, row counts). Used by the
    multi-crop list-forward so norms/FFN run on one fused batch
    (reference utils/utils.py:14-26).
    """
    shapes = [t.shape for t in tensors]
    counts = [t.shape[:-1].numel() for t in tensors]
    flat = torch.cat([t.reshape(-1, t.shape[-1]) for t in tensors], dim=0)
    return flat, list(shapes), counts


def uncat_with_shapes(flat: torch.Tensor, shapes: List[torch.Size], counts: List[int]) -> List[torch.Tensor]:
    """Inverse of cat_keep_shapes (reference utils/utils.py:27-35)."""
    splits = torch.split(flat, counts, dim=0)
    return [s.reshape(*shape[:-1], flat.shape[-1]) for s, shape in zip(splits, shapes)]


def fix_random_seeds(seed: int = 31) -> None:
    torch.manual_seed(seed)
    np.random.seed(seed)
    random.seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def count_parameters(module: torch.nn.Module) -> int:
    return sum(p.numel() for p in module.parameters())
