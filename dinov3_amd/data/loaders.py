"""Dataset-string parsing + DataLoader construction.

Parity: dinov3_jax/data/loaders.py:22-217 ("ImageNet:split=TRAIN" strings,
SamplerType selection, torch DataLoader wrap). Unlike the reference (which
forces num_workers=0), workers + pinned memory are on by default on GPU runs.
"""

from __future__ import annotations

import logging
from enum import Enum
from typing import Callable, Optional

import torch

from . import datasets as ds
from .samplers import EpochSampler, InfiniteSampler, ShardedInfiniteSampler

logger = logging.getLogger("dinov3")


class SamplerType(Enum):
    EPOCH = 0
    INFINITE = 1
    SHARDED_INFINITE = 2


_DATASETS = {
    "ImageNet": ds.ImageNet,
    "ImageNet22k": ds.ImageNet22k,
    "ADE20K": ds.ADE20K,
    "CocoCaptions": ds.CocoCaptions,
    "Synthetic": ds.SyntheticDataset,
}


def _parse_dataset_str(dataset_str: str):
    tokens = dataset_str.split(":")
    name = tokens[0]
    kwargs = {}
    for token in tokens[1:]:
        key, _, value = token.partition("=")
        assert key in ("root", "extra", "split", "length"), f"unsupported dataset arg {key}"
        if key == "split":
            kwargs["split"] = ds.Split[value]
        elif key == "length":
            kwargs["length"] = int(value)
        else:
            kwargs[key] = value
    if name not in _DATASETS:
        raise ValueError(f'Unsupported dataset "{name}"')
    return _DATASETS[name], kwargs


def make_dataset(*, dataset_str: str, transform: Optional[Callable] = None,
                 target_transform: Optional[Callable] = None):
    logger.info('using dataset: "%s"', dataset_str)
    cls, kwargs = _parse_dataset_str(dataset_str)
    dataset = cls(transform=transform, target_transform=target_transform, **kwargs)
    logger.info("# of dataset samples: %d", len(dataset))
    return dataset


def _make_sampler(*, dataset, type: Optional[SamplerType], shuffle: bool, seed: int, advance: int):
    sample_count = len(dataset)
    if type == SamplerType.EPOCH:
        return EpochSampler(size=sample_count, sample_count=sample_count, shuffle=shuffle, seed=seed,
                            advance=advance)
    if type == SamplerType.INFINITE:
        return InfiniteSampler(sample_count=sample_count, shuffle=shuffle, seed=seed, advance=advance)
    if type == SamplerType.SHARDED_INFINITE:
        return ShardedInfiniteSampler(sample_count=sample_count, shuffle=shuffle, seed=seed, advance=advance)
    return None


class CombinedDataLoader:
    """Sample batches from several loaders with given probabilities.

    The reference references a missing `CombineDataLoader` for its
    multi-resolution crop schedules (train.py:763) — this is
    the working equivalent: every __iter__ draws a loader by ratio, yielding
    its next batch (each batch is internally one resolution).
    """

    def __init__(self, loaders, ratios=None, seed: int = 0):
        import numpy as np

        self.loaders = list(loaders)
        if ratios is None:
            ratios = [1.0] * len(self.loaders)
        total = float(sum(ratios))
        self.ratios = [r / total for r in ratios]
        self._rng = np.random.default_rng(seed)

    def __iter__(self):
        import numpy as np

        iters = [iter(dl) for dl in self.loaders]
        while True:
            choice = int(self._rng.choice(len(iters), p=self.ratios))
            try:
                yield next(iters[choice])
            except StopIteration:
                iters[choice] = iter(self.loaders[choice])
                yield next(iters[choice])

    def __len__(self):
        return sum(len(dl) for dl in self.loaders)

    def set_epoch(self, epoch: int) -> None:
        for dl in self.loaders:
            sampler = getattr(dl, "sampler", None)
            if sampler is not None and hasattr(sampler, "set_epoch"):
                sampler.set_epoch(epoch)


def make_data_loader(
    *,
    dataset,
    batch_size: int,
    num_workers: int = 0,
    shuffle: bool = True,
    seed: int = 0,
    sampler_type: Optional[SamplerType] = SamplerType.EPOCH,
    sampler_advance: int = 0,
    drop_last: bool = True,
    persistent_workers: bool = False,
    pin_memory: Optional[bool] = None,
    collate_fn: Optional[Callable] = None,
):
    sampler = _make_sampler(dataset=dataset, type=sampler_type, shuffle=shuffle, seed=seed,
                            advance=sampler_advance)
    if pin_memory is None:
        pin_memory = torch.cuda.is_available()
    loader = torch.utils.data.DataLoader(
        dataset,
        sampler=sampler,
        batch_size=batch_size,
        num_workers=num_workers,
        pin_memory=pin_memory,
        drop_last=drop_last,
        persistent_workers=persistent_workers and num_workers > 0,
        collate_fn=collate_fn,
    )
    logger.info("data loader: batch_size %d, workers %d, pin %s", batch_size, num_workers, pin_memory)
    return loader
