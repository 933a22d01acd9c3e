"""On-GPU augmentation training pipeline.

CPU multi-crop augmentation costs ~25 ms/image/core — feeding one MI355X at
~450 img/s needs ~12 dataloader workers. This pipeline decouples throughput
from host cores: workers only DECODE (and resize to one canonical size),
batches go H2D once as uint8, and the full DINO multi-crop recipe runs as
batched tensor ops on the training GPU (data/gpu_augment.py). iBOT masks stay
on the host (the BEiT block-mask generator is inherently sequential) and ride
along with the batch.

Output batches have exactly the collate_data_and_cast contract
(crop-major `collated_global_crops` / `collated_local_crops` + mask fields),
so SSLMetaArch consumes them unchanged.

Deviation from the per-sample CPU path, documented for parity review: crops
are taken from the canonical-size decode (default 256px shortest-side resize
+ center crop) rather than the original-resolution image, and resampling is
bilinear rather than antialiased bicubic.
"""

from __future__ import annotations

import logging
from typing import Iterator, Optional

import torch
import torch.nn.functional as F

from .collate import build_mask_batch
from .gpu_augment import GpuDataAugmentationDINO
from .masking import MaskingGenerator

logger = logging.getLogger("dinov3")


class CanonicalDecode:
    """Worker-side transform: float CHW in [0,1] -> uint8 CHW at size²
    (shortest-side resize + center crop)."""

    def __init__(self, size: int = 256):
        self.size = size

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        c, h, w = img.shape
        s = self.size / min(h, w)
        nh, nw = max(self.size, int(round(h * s))), max(self.size, int(round(w * s)))
        if (nh, nw) != (h, w):
            img = F.interpolate(img.unsqueeze(0), size=(nh, nw), mode="bilinear",
                                align_corners=False).squeeze(0)
        top, left = (nh - self.size) // 2, (nw - self.size) // 2
        img = img[:, top: top + self.size, left: left + self.size]
        return (img.clamp(0, 1) * 255.0).to(torch.uint8)


def raw_collate(samples):
    """[(uint8 CHW, target)] -> (uint8 [B,3,S,S], ignored)."""
    return torch.stack([s[0] for s in samples]), None


class GpuAugmentPipeline:
    """Iterates (raw uint8 batches from `raw_loader`) -> full collated DINO
    batches living on `device`."""

    def __init__(self, raw_loader, cfg, device: torch.device, dtype: torch.dtype):
        self.raw_loader = raw_loader
        self.device = device
        self.dtype = dtype
        crops = cfg.crops
        self.aug = GpuDataAugmentationDINO(
            global_crops_scale=tuple(crops.global_crops_scale),
            local_crops_scale=tuple(crops.local_crops_scale),
            local_crops_number=crops.local_crops_number,
            global_crops_size=crops.global_crops_size,
            local_crops_size=crops.local_crops_size,
            mean=tuple(crops.rgb_mean), std=tuple(crops.rgb_std),
        )
        p = cfg.student.patch_size
        gs = crops.global_crops_size
        self.n_tokens = (gs // p) ** 2
        self.mask_generator = MaskingGenerator(
            input_size=(gs // p, gs // p), max_num_patches=int(0.5 * self.n_tokens))
        self.mask_ratio_tuple = tuple(cfg.ibot.mask_ratio_min_max)
        self.mask_probability = cfg.ibot.mask_sample_probability
        self.random_circular_shift = cfg.ibot.mask_random_circular_shift

    @property
    def sampler(self):
        return self.raw_loader.sampler

    def __len__(self):
        return len(self.raw_loader)

    def __iter__(self) -> Iterator[dict]:
        for imgs, _ in self.raw_loader:
            imgs = imgs.to(self.device, non_blocking=True)
            crops = self.aug(imgs)
            masks = build_mask_batch(
                crops["global_crops"].shape[0], self.n_tokens, self.mask_generator,
                self.mask_ratio_tuple, self.mask_probability,
                random_circular_shift=self.random_circular_shift,
            )
            yield {
                "collated_global_crops": crops["global_crops"].to(self.dtype),
                "collated_local_crops": crops["local_crops"].to(self.dtype),
                **{k: (v.to(self.device, non_blocking=True)
                       if isinstance(v, torch.Tensor) else v)
                   for k, v in masks.items()},
            }


def build_gpu_augment_pipeline_from_cfg(cfg, device: torch.device, dtype: torch.dtype,
                                        canonical_size: int = 256,
                                        sampler_advance: int = 0):
    """Raw-decode DataLoader (workers) + on-device augmentation pipeline."""
    from .loaders import SamplerType, make_data_loader, make_dataset

    dataset = make_dataset(
        dataset_str=cfg.train.dataset_path,
        transform=CanonicalDecode(canonical_size),
        target_transform=lambda _: (),
    )
    raw_loader = make_data_loader(
        dataset=dataset,
        batch_size=cfg.train.batch_size_per_gpu,
        num_workers=cfg.train.num_workers,
        shuffle=True,
        seed=cfg.train.seed,
        sampler_type=SamplerType.EPOCH,
        sampler_advance=sampler_advance,
        drop_last=True,
        collate_fn=raw_collate,
    )
    return GpuAugmentPipeline(raw_loader, cfg, device, dtype)
