"""Block-wise iBOT mask generator (BEiT-style).

Parity: dinov3_jax/data/masking.py:14-100 — random rectangles of bounded
aspect/area until the target count, then random fill/trim to the exact count.
"""

from __future__ import annotations

import math
import random
from typing import Optional, Tuple, Union

import numpy as np


class MaskingGenerator:
    def __init__(
        self,
        input_size: Union[int, Tuple[int, int]],
        num_masking_patches: Optional[int] = None,
        min_num_patches: int = 4,
        max_num_patches: Optional[int] = None,
        min_aspect: float = 0.3,
        max_aspect: Optional[float] = None,
    ):
        if not isinstance(input_size, tuple):
            input_size = (input_size, input_size)
        self.height, self.width = input_size
        self.num_masking_patches = num_masking_patches
        self.min_num_patches = min_num_patches
        self.max_num_patches = num_masking_patches if max_num_patches is None else max_num_patches
        max_aspect = max_aspect or 1 / min_aspect
        self.log_aspect_ratio = (math.log(min_aspect), math.log(max_aspect))

    def get_shape(self) -> Tuple[int, int]:
        return self.height, self.width

    def _add_block(self, mask: np.ndarray, max_mask_patches: int) -> int:
        """Try (up to 10 times) to add one new rectangle; returns #new cells."""
        for _ in range(10):
            target_area = random.uniform(self.min_num_patches, max_mask_patches)
            aspect = math.exp(random.uniform(*self.log_aspect_ratio))
            h = int(round(math.sqrt(target_area * aspect)))
            w = int(round(math.sqrt(target_area / aspect)))
            if w < self.width and h < self.height:
                top = random.randint(0, self.height - h)
                left = random.randint(0, self.width - w)
                region = mask[top: top + h, left: left + w]
                new_cells = h * w - int(region.sum())
                if 0 < new_cells <= max_mask_patches:
                    region |= True
                    return new_cells
        return 0

    def _fill_to_exact(self, mask: np.ndarray, target: int) -> np.ndarray:
        flat = mask.flatten()
        have = int(flat.sum())
        if have < target:
            candidates = np.where(~flat)[0]
            extra = np.random.choice(candidates, size=target - have, replace=False)
            flat[extra] = True
        elif have > target:
            on = np.where(flat)[0]
            drop = np.random.choice(on, size=have - target, replace=False)
            flat[drop] = False
        return flat.reshape(mask.shape)

    def __call__(self, num_masking_patches: int = 0) -> np.ndarray:
        mask = np.zeros(self.get_shape(), dtype=bool)
        count = 0
        while count < num_masking_patches:
            budget = min(num_masking_patches - count, self.max_num_patches or num_masking_patches)
            delta = self._add_block(mask, budget)
            if delta == 0:
                break
            count += delta
        return self._fill_to_exact(mask, num_masking_patches)
