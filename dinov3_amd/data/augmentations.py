"""DINO multi-crop augmentation pipeline, torch-native.

Parity: dinov3_jax/data/augmentations.py:23-230 — 2 global RandomResizedCrops
(blur p=1 on #1; blur p=.1 + solarize p=.2 on #2), N local crops (blur p=.5),
shared/unshared color jitter (brightness .4, contrast .4, saturation .2, hue
.1 @ p=.8; grayscale p=.2), ImageNet normalize, optional gram-teacher crops
and local-crops-as-subwindows-of-global.

Operates on float32 CHW tensors in [0,1] (see transforms.py).
"""

from __future__ import annotations

import logging
import random
from typing import Optional, Sequence

import numpy as np
import torch

from . import transforms as T

logger = logging.getLogger("dinov3")


class DataAugmentationDINO:
    def __init__(
        self,
        global_crops_scale: Sequence[float],
        local_crops_scale: Sequence[float],
        local_crops_number: int,
        global_crops_size: int = 224,
        local_crops_size: int = 96,
        gram_teacher_crops_size: Optional[int] = None,
        gram_teacher_no_distortions: bool = False,
        teacher_no_color_jitter: bool = False,
        local_crops_subset_of_global_crops: bool = False,
        patch_size: int = 16,
        share_color_jitter: bool = False,
        horizontal_flips: bool = True,
        mean: Sequence[float] = (0.485, 0.456, 0.406),
        std: Sequence[float] = (0.229, 0.224, 0.225),
    ):
        self.global_crops_scale = tuple(global_crops_scale)
        self.local_crops_scale = tuple(local_crops_scale)
        self.local_crops_number = local_crops_number
        self.global_crops_size = global_crops_size
        self.local_crops_size = local_crops_size
        self.gram_teacher_crops_size = gram_teacher_crops_size
        self.gram_teacher_no_distortions = gram_teacher_no_distortions
        self.teacher_no_color_jitter = teacher_no_color_jitter
        self.local_crops_subset_of_global_crops = local_crops_subset_of_global_crops
        self.patch_size = patch_size
        self.share_color_jitter = share_color_jitter
        self.horizontal_flips = horizontal_flips
        self.mean = tuple(mean)
        self.std = tuple(std)

    # --- building blocks -------------------------------------------------
    def _geom_global(self, image: torch.Tensor) -> torch.Tensor:
        crop = T.random_resized_crop(image, self.global_crops_size, self.global_crops_scale)
        if self.horizontal_flips and random.random() < 0.5:
            crop = T.hflip(crop)
        return crop

    def _geom_local(self, image: torch.Tensor) -> torch.Tensor:
        crop = T.random_resized_crop(image, self.local_crops_size, self.local_crops_scale)
        if self.horizontal_flips and random.random() < 0.5:
            crop = T.hflip(crop)
        return crop

    def _color(self, img: torch.Tensor) -> torch.Tensor:
        if random.random() < 0.8:
            img = T.color_jitter(img, 0.4, 0.4, 0.2, 0.1)
        if random.random() < 0.2:
            img = T.rgb_to_grayscale(img)
        return img

    def _blur(self, img: torch.Tensor, p: float) -> torch.Tensor:
        if random.random() < p:
            img = T.gaussian_blur(img, sigma=random.uniform(0.1, 2.0))
        return img

    def _normalize(self, img: torch.Tensor) -> torch.Tensor:
        return T.normalize(img, self.mean, self.std)

    # --- main ------------------------------------------------------------
    def __call__(self, image: torch.Tensor) -> dict:
        output: dict = {"weak_flag": True}
        if self.share_color_jitter:
            image = self._color(image)

        im1_base = self._geom_global(image)
        g1_transf = im1_base if self.share_color_jitter else self._color(im1_base)
        g1_transf = self._blur(g1_transf, p=1.0)
        global_crop_1 = self._normalize(g1_transf)

        im2_base = self._geom_global(image)
        g2_transf = im2_base if self.share_color_jitter else self._color(im2_base)
        g2_transf = self._blur(g2_transf, p=0.1)
        if random.random() < 0.2:
            g2_transf = T.solarize(g2_transf, 0.5)
        global_crop_2 = self._normalize(g2_transf)

        output["global_crops"] = [global_crop_1, global_crop_2]
        if self.teacher_no_color_jitter:
            output["global_crops_teacher"] = [self._normalize(im1_base), self._normalize(im2_base)]
        else:
            output["global_crops_teacher"] = [global_crop_1, global_crop_2]

        if self.gram_teacher_crops_size is not None:
            if self.gram_teacher_no_distortions:
                gram_1 = self._normalize(T.resize(im1_base, self.gram_teacher_crops_size))
                gram_2 = self._normalize(T.resize(im2_base, self.gram_teacher_crops_size))
            else:
                gram_1 = self._normalize(T.resize(g1_transf, self.gram_teacher_crops_size))
                gram_2 = self._normalize(T.resize(g2_transf, self.gram_teacher_crops_size))
            output["gram_teacher_crops"] = [gram_1, gram_2]

        if self.local_crops_subset_of_global_crops:
            half = self.local_crops_number // 2
            bases = [im1_base] * half + [im2_base] * (self.local_crops_number - half)
            local_crops, offsets = [], []
            gs, ls, p = self.global_crops_size, self.local_crops_size, self.patch_size
            for base in bases:
                img = self._normalize(self._blur(self._color(base), p=0.5))
                rx, ry = np.random.randint(0, (gs - ls) // p, 2) * p
                local_crops.append(img[:, rx: rx + ls, ry: ry + ls])
                offsets.append((int(rx), int(ry)))
            output["local_crops"] = local_crops
            output["offsets"] = offsets
        else:
            output["local_crops"] = [
                self._normalize(self._blur(self._color(self._geom_local(image)), p=0.5))
                for _ in range(self.local_crops_number)
            ]
            output["offsets"] = ()
        return output
