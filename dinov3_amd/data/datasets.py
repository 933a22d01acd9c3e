"""Datasets. The reference ships its I/O bypassed — decoders return random
224x224 images and random targets (dinov3_jax/data/datasets/decoders.py:28-45,
image_net.py:170-195) so training runs with no dataset on disk. We keep that
capability as an explicit synthetic decode path (the only one usable in this
offline environment) behind the same dataset names/split API.
"""

from __future__ import annotations

import logging
from enum import Enum
from typing import Any, Callable, Optional

import numpy as np
import torch

logger = logging.getLogger("dinov3")


class Split(Enum):
    TRAIN = "TRAIN"
    VAL = "VAL"
    TEST = "TEST"


def random_image(height: int = 224, width: int = 224, seed: Optional[int] = None) -> torch.Tensor:
    """float32 CHW in [0,1] (the synthetic decoder)."""
    rng = np.random.default_rng(seed)
    arr = rng.integers(0, 256, size=(3, height, width), dtype=np.uint8)
    return torch.from_numpy(arr).float() / 255.0


class ExtendedVisionDataset(torch.utils.data.Dataset):
    """Base: decode -> transform, with (image, target) tuple output."""

    def __init__(self, transform: Optional[Callable] = None, target_transform: Optional[Callable] = None):
        self.transform = transform
        self.target_transform = target_transform

    def get_image_data(self, index: int) -> Optional[bytes]:
        return None  # synthetic mode

    def get_target(self, index: int) -> Any:
        return None

    def decode_image(self, data: Optional[bytes], index: int) -> torch.Tensor:
        return random_image()

    def __getitem__(self, index: int):
        image = self.decode_image(self.get_image_data(index), index)
        target = self.get_target(index)
        if self.transform is not None:
            image = self.transform(image)
        if self.target_transform is not None:
            target = self.target_transform(target)
        return image, target


_SPLIT_LENGTHS = {
    "ImageNet": {Split.TRAIN: 1_281_167, Split.VAL: 50_000, Split.TEST: 100_000},
    "ImageNet22k": {Split.TRAIN: 11_797_647, Split.VAL: 561_052, Split.TEST: 561_052},
    "ADE20K": {Split.TRAIN: 20_210, Split.VAL: 2_000, Split.TEST: 3_352},
    "CocoCaptions": {Split.TRAIN: 118_287, Split.VAL: 5_000, Split.TEST: 40_670},
}


class _SyntheticSplitDataset(ExtendedVisionDataset):
    NAME = "Synthetic"
    NUM_CLASSES = 1000

    def __init__(self, split: Split = Split.TRAIN, root: str = "", transform=None, target_transform=None,
                 length: Optional[int] = None):
        super().__init__(transform, target_transform)
        self.split = split
        self.root = root
        default = _SPLIT_LENGTHS.get(self.NAME, {}).get(split, 10_000)
        self._length = length if length is not None else default

    def get_target(self, index: int) -> int:
        return int(np.random.randint(self.NUM_CLASSES))

    def __len__(self) -> int:
        return self._length


class ImageNet(_SyntheticSplitDataset):
    NAME = "ImageNet"
    NUM_CLASSES = 1000


class ImageNet22k(_SyntheticSplitDataset):
    NAME = "ImageNet22k"
    NUM_CLASSES = 21_841


class ADE20K(_SyntheticSplitDataset):
    NAME = "ADE20K"
    NUM_CLASSES = 150


class CocoCaptions(_SyntheticSplitDataset):
    NAME = "CocoCaptions"
    NUM_CLASSES = 0

    def get_target(self, index: int) -> str:
        return ""


class SyntheticDataset(_SyntheticSplitDataset):
    """Explicit synthetic dataset with configurable size/resolution."""

    NAME = "Synthetic"

    def __init__(self, split: Split = Split.TRAIN, root: str = "", transform=None, target_transform=None,
                 length: int = 10_000, height: int = 224, width: int = 224):
        super().__init__(split, root, transform, target_transform, length=length)
        self.height = height
        self.width = width

    def decode_image(self, data, index: int) -> torch.Tensor:
        return random_image(self.height, self.width)
