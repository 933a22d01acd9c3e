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

    def __init__(self, split: Split = Split.TRAIN, root: str = "", extra: str = "",
                 transform=None, target_transform=None, length: Optional[int] = None):
        super().__init__(transform, target_transform)
        self.split = split
        self.root = root
        self.extra = extra
        default = _SPLIT_LENGTHS.get(self.NAME, {}).get(split, 10_000)
        self._length = length if length is not None else default

    def get_target(self, index: int) -> int:
        return int(np.random.randint(self.NUM_CLASSES))

    def __len__(self) -> int:
        return self._length


class _RealReaderMixin:
    """Datasets with a real on-disk reader when root+extra are configured
    (synthetic decode otherwise — the offline default). `_make_reader` returns
    an object with get_image_data/get_target/__len__."""

    _reader = None

    def _attach_reader(self, length_override: Optional[int]) -> None:
        if self.root and self.extra:
            self._reader = self._make_reader()
            self._length = length_override if length_override is not None else len(self._reader)

    def get_image_data(self, index: int) -> Optional[bytes]:
        return self._reader.get_image_data(index) if self._reader is not None else None

    def get_target(self, index: int):
        if self._reader is not None:
            return self._reader.get_target(index)
        return super().get_target(index)

    def get_targets(self) -> Optional[np.ndarray]:
        return self._reader.get_targets() if self._reader is not None else None

    def decode_image(self, data: Optional[bytes], index: int) -> torch.Tensor:
        if data is not None:
            from .readers import decode_image_bytes

            return decode_image_bytes(data)
        return random_image()


class ImageNet(_RealReaderMixin, _SyntheticSplitDataset):
    """Real mode: npy entries/class-ids index over an image tree (reference
    dinov3_jax/data/datasets/image_net.py:27-337, I/O un-bypassed here)."""

    NAME = "ImageNet"
    NUM_CLASSES = 1000

    def __init__(self, split: Split = Split.TRAIN, root: str = "", extra: str = "",
                 transform=None, target_transform=None, length: Optional[int] = None):
        super().__init__(split, root, extra, transform, target_transform, length)
        self._attach_reader(length)

    def _make_reader(self):
        from .readers import ImageNetIndexReader

        return ImageNetIndexReader(self.root, self.extra, self.split.value.lower())


class ImageNet22k(_RealReaderMixin, _SyntheticSplitDataset):
    """Real mode: per-class tarballs + block-offset entries.npy (reference
    image_net_22k.py, mmap cache + gzip members honored)."""

    NAME = "ImageNet22k"
    NUM_CLASSES = 21_841

    def __init__(self, split: Split = Split.TRAIN, root: str = "", extra: str = "",
                 transform=None, target_transform=None, length: Optional[int] = None,
                 mmap_cache_size: int = 16):
        super().__init__(split, root, extra, transform, target_transform, length)
        self._mmap_cache_size = mmap_cache_size
        self._attach_reader(length)

    def _make_reader(self):
        from .readers import ImageNet22kTarballReader

        return ImageNet22kTarballReader(self.root, self.extra, self._mmap_cache_size)


class ADE20K(_RealReaderMixin, _SyntheticSplitDataset):
    """Real mode: split txt listing + images/ + annotations/ PNG masks
    (reference ade20k.py:32-102, I/O un-bypassed here)."""

    NAME = "ADE20K"
    NUM_CLASSES = 150

    def __init__(self, split: Split = Split.TRAIN, root: str = "", extra: str = "",
                 transform=None, target_transform=None, length: Optional[int] = None):
        super().__init__(split, root, extra, transform, target_transform, length)
        if root:  # ADE20K has no separate "extra" dir — root alone enables it
            self.extra = root
        self._attach_reader(length)

    def _make_reader(self):
        from .readers import ADE20KReader

        return ADE20KReader(self.root, self.split.value.lower())


class CocoCaptions(_RealReaderMixin, _SyntheticSplitDataset):
    """Real mode: COCO caption json + image dirs (reference
    coco_captions.py:28-102); target = a random caption of the image."""

    NAME = "CocoCaptions"
    NUM_CLASSES = 0

    def __init__(self, split: Split = Split.TRAIN, root: str = "", extra: str = "",
                 transform=None, target_transform=None, length: Optional[int] = None):
        super().__init__(split, root, extra, transform, target_transform, length)
        if root:
            self.extra = root
        self._attach_reader(length)

    def _make_reader(self):
        from .readers import CocoCaptionsReader

        return CocoCaptionsReader(self.root, self.split.value.lower())

    def get_target(self, index: int) -> str:
        if self._reader is not None:
            return self._reader.get_target(index)
        return ""


class SyntheticDataset(_SyntheticSplitDataset):
    """Explicit synthetic dataset with configurable size/resolution."""

    NAME = "Synthetic"

    def __init__(self, split: Split = Split.TRAIN, root: str = "", extra: str = "",
                 transform=None, target_transform=None,
                 length: int = 10_000, height: int = 224, width: int = 224):
        super().__init__(split, root, extra, transform, target_transform, length=length)
        self.height = height
        self.width = width

    def decode_image(self, data, index: int) -> torch.Tensor:
        return random_image(self.height, self.width)
