"""Batched on-device DINO multi-crop augmentation.

The per-sample CPU pipeline (augmentations.py) costs ~25 ms/image on one
core — a ViT-L node at ~430 img/s/GPU would need ~12 dataloader workers per
GPU to keep up. This module runs the same recipe as batched tensor ops on
the training GPU instead: decoded uint8 batches go H2D once, then every
crop group is produced with a handful of batched kernels (one grid_sample
per crop group + vectorized jitter/blur/solarize), so augmentation scales
with GPU bandwidth rather than host cores.

Semantics follow DataAugmentationDINO (reference dinov3_jax/data/
augmentations.py:23-230): 2 global crops (scale 0.32-1.0) — #1 always
blurred, #2 blur p=.1 + solarize p=.2 — N local crops (0.05-0.32), hflip
p=.5, shared/unshared color jitter (brightness .4, contrast .4,
saturation .2, hue .1, p=.8), grayscale p=.2, ImageNet normalize. Two
deliberate deviations, documented for parity review: resizing is bilinear
grid_sample (the CPU path uses antialiased bicubic), and the blur kernel
size is capped at GAUSS_KSIZE taps (per-image sigma is exact).

Randomness is drawn on the host (python `random`, like the CPU pipeline)
into per-image parameter tensors; all heavy math is batched on the device.
Works identically on CPU tensors, which is how the equivalence tests run.
"""

from __future__ import annotations

import random
from typing import Dict, Tuple

import torch
import torch.nn.functional as F

from .transforms import sample_rrc_box

IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)
GAUSS_KSIZE = 9  # max taps per axis; matches common GPU DINO pipelines


# --------------------------- batched primitives ---------------------------


def batched_rrc_flip(imgs: torch.Tensor, boxes: torch.Tensor, flips: torch.Tensor,
                     size: int) -> torch.Tensor:
    """Crop box + resize + optional hflip for every image in one grid_sample.

    imgs: [B,3,H,W] float; boxes: [B,4] (top, left, h, w) in pixels;
    flips: [B] bool. Returns [B,3,size,size].
    """
    B, _, H, W = imgs.shape
    top, left, ch, cw = boxes.unbind(dim=1)
    # affine theta mapping output [-1,1]^2 to the box (align_corners=False)
    cx = (left + cw * 0.5) / W * 2.0 - 1.0
    cy = (top + ch * 0.5) / H * 2.0 - 1.0
    sx = cw.float() / W
    sy = ch.float() / H
    sx = torch.where(flips, -sx, sx)
    theta = torch.zeros(B, 2, 3, device=imgs.device, dtype=torch.float32)
    theta[:, 0, 0] = sx
    theta[:, 1, 1] = sy
    theta[:, 0, 2] = cx
    theta[:, 1, 2] = cy
    grid = F.affine_grid(theta, (B, 3, size, size), align_corners=False)
    return F.grid_sample(imgs, grid, mode="bilinear", padding_mode="border",
                         align_corners=False).clamp(0.0, 1.0)


def batched_grayscale(imgs: torch.Tensor) -> torch.Tensor:
    w = imgs.new_tensor([0.2989, 0.587, 0.114]).view(1, 3, 1, 1)
    return (imgs * w).sum(dim=1, keepdim=True).expand_as(imgs)


def batched_brightness(imgs: torch.Tensor, f: torch.Tensor) -> torch.Tensor:
    return (imgs * f.view(-1, 1, 1, 1)).clamp(0.0, 1.0)


def batched_contrast(imgs: torch.Tensor, f: torch.Tensor) -> torch.Tensor:
    mean = batched_grayscale(imgs).mean(dim=(1, 2, 3), keepdim=True)
    fv = f.view(-1, 1, 1, 1)
    return (imgs * fv + mean * (1.0 - fv)).clamp(0.0, 1.0)


def batched_saturation(imgs: torch.Tensor, f: torch.Tensor) -> torch.Tensor:
    g = batched_grayscale(imgs)
    fv = f.view(-1, 1, 1, 1)
    return (imgs * fv + g * (1.0 - fv)).clamp(0.0, 1.0)


def batched_hue(imgs: torch.Tensor, shift: torch.Tensor) -> torch.Tensor:
    """Branch-free batched HSV hue rotation; shift [B] in [-0.5, 0.5] turns.
    Same k-formula as transforms.adjust_hue."""
    r, g, b = imgs.unbind(dim=1)
    maxc, _ = imgs.max(dim=1)
    minc, _ = imgs.min(dim=1)
    v = maxc
    deltac = maxc - minc
    s = deltac / maxc.clamp_min(1e-8)
    dz = deltac.clamp_min(1e-8)
    rc = (maxc - r) / dz
    gc = (maxc - g) / dz
    bc = (maxc - b) / dz
    h = torch.where(r == maxc, bc - gc, torch.where(g == maxc, 2.0 + rc - bc, 4.0 + gc - rc))
    h6 = (h % 6.0) + shift.view(-1, 1, 1) * 6.0
    chans = []
    for n in (5.0, 3.0, 1.0):
        k = (n + h6) % 6.0
        chans.append(v - v * s * torch.clamp(torch.minimum(k, 4.0 - k), 0.0, 1.0))
    return torch.stack(chans, dim=1).clamp(0.0, 1.0)


def batched_gaussian_blur(imgs: torch.Tensor, sigma: torch.Tensor,
                          apply: torch.Tensor) -> torch.Tensor:
    """Separable blur with per-image sigma via grouped conv (one call per
    axis). sigma [B]; apply [B] bool — non-applied rows get identity kernels."""
    B, C, H, W = imgs.shape
    k = min(GAUSS_KSIZE, (min(H, W) // 2) * 2 + 1)
    x = torch.arange(k, device=imgs.device, dtype=torch.float32) - k // 2
    sig = sigma.to(imgs.device).float().clamp_min(1e-3).view(B, 1)
    kern = torch.exp(-0.5 * (x.view(1, k) / sig) ** 2)
    kern = kern / kern.sum(dim=1, keepdim=True)
    ident = torch.zeros(k, device=imgs.device)
    ident[k // 2] = 1.0
    kern = torch.where(apply.view(B, 1).to(imgs.device), kern, ident.view(1, k))
    kx = kern.repeat_interleave(C, dim=0).view(B * C, 1, 1, k)
    ky = kx.view(B * C, 1, k, 1)
    pad = k // 2
    out = imgs.reshape(1, B * C, H, W)
    out = F.conv2d(out, kx, padding=(0, pad), groups=B * C)
    out = F.conv2d(out, ky, padding=(pad, 0), groups=B * C)
    return out.reshape(B, C, H, W)


def batched_solarize(imgs: torch.Tensor, apply: torch.Tensor,
                     threshold: float = 0.5) -> torch.Tensor:
    sol = torch.where(imgs >= threshold, 1.0 - imgs, imgs)
    return torch.where(apply.view(-1, 1, 1, 1).to(imgs.device), sol, imgs)


def batched_color_jitter(imgs: torch.Tensor, params: Dict[str, torch.Tensor]) -> torch.Tensor:
    """Apply jitter ops in the per-batch shuffled order of params["order"].
    Images whose `apply` flag is off get factor 1 / shift 0 (identity)."""
    for kind in params["order"]:
        if kind == "b":
            imgs = batched_brightness(imgs, params["b"])
        elif kind == "c":
            imgs = batched_contrast(imgs, params["c"])
        elif kind == "s":
            imgs = batched_saturation(imgs, params["s"])
        else:
            imgs = batched_hue(imgs, params["h"])
    gray = batched_grayscale(imgs)
    return torch.where(params["gray"].view(-1, 1, 1, 1).to(imgs.device), gray, imgs)


# ------------------------------ the pipeline ------------------------------


class GpuDataAugmentationDINO:
    """Batched DINO multi-crop on device. __call__ takes decoded images
    [B,3,H,W] (uint8 or float in [0,1]) and returns crop-major tensors:
    global_crops [2B,3,gs,gs], local_crops [n_local*B,3,ls,ls] — the layout
    collate_data_and_cast produces (crop index major, then sample index)."""

    def __init__(self, global_crops_scale=(0.32, 1.0), local_crops_scale=(0.05, 0.32),
                 local_crops_number: int = 8, global_crops_size: int = 224,
                 local_crops_size: int = 96,
                 jitter=(0.4, 0.4, 0.2, 0.1), jitter_prob: float = 0.8,
                 gray_prob: float = 0.2, mean=IMAGENET_MEAN, std=IMAGENET_STD):
        self.global_crops_scale = tuple(global_crops_scale)
        self.local_crops_scale = tuple(local_crops_scale)
        self.local_crops_number = local_crops_number
        self.global_crops_size = global_crops_size
        self.local_crops_size = local_crops_size
        self.jitter = jitter
        self.jitter_prob = jitter_prob
        self.gray_prob = gray_prob
        self.mean = mean
        self.std = std

    # -- host-side RNG --
    def _sample_boxes(self, B: int, H: int, W: int, scale) -> Tuple[torch.Tensor, torch.Tensor]:
        boxes = torch.tensor([sample_rrc_box(H, W, scale) for _ in range(B)],
                             dtype=torch.float32)
        flips = torch.tensor([random.random() < 0.5 for _ in range(B)])
        return boxes, flips

    def _sample_jitter(self, B: int) -> Dict[str, torch.Tensor]:
        bj, cj, sj, hj = self.jitter
        on = [random.random() < self.jitter_prob for _ in range(B)]

        def factor(strength):
            return torch.tensor([
                random.uniform(max(0.0, 1 - strength), 1 + strength) if o else 1.0
                for o in on])

        order = ["b", "c", "s", "h"]
        random.shuffle(order)
        return {
            "order": order,
            "b": factor(bj),
            "c": factor(cj),
            "s": factor(sj),
            "h": torch.tensor([random.uniform(-hj, hj) if o else 0.0 for o in on]),
            "gray": torch.tensor([random.random() < self.gray_prob for _ in range(B)]),
        }

    def _blur_params(self, B: int, prob: float) -> Tuple[torch.Tensor, torch.Tensor]:
        apply = torch.tensor([random.random() < prob for _ in range(B)])
        sigma = torch.tensor([random.uniform(0.1, 2.0) for _ in range(B)])
        return sigma, apply

    def _finish(self, crops: torch.Tensor) -> torch.Tensor:
        mean = crops.new_tensor(self.mean).view(1, 3, 1, 1)
        std = crops.new_tensor(self.std).view(1, 3, 1, 1)
        return (crops - mean) / std

    def _one_group(self, imgs: torch.Tensor, size: int, scale, blur_prob: float,
                   solarize_prob: float = 0.0) -> torch.Tensor:
        B, _, H, W = imgs.shape
        boxes, flips = self._sample_boxes(B, H, W, scale)
        crops = batched_rrc_flip(imgs, boxes.to(imgs.device), flips.to(imgs.device), size)
        jitter = {k: v.to(imgs.device) if torch.is_tensor(v) else v
                  for k, v in self._sample_jitter(B).items()}
        crops = batched_color_jitter(crops, jitter)
        sigma, apply = self._blur_params(B, blur_prob)
        crops = batched_gaussian_blur(crops, sigma, apply)
        if solarize_prob > 0:
            sol = torch.tensor([random.random() < solarize_prob for _ in range(B)])
            crops = batched_solarize(crops, sol)
        return self._finish(crops)

    def __call__(self, imgs: torch.Tensor) -> Dict[str, torch.Tensor]:
        if imgs.dtype == torch.uint8:
            imgs = imgs.float() / 255.0
        g1 = self._one_group(imgs, self.global_crops_size, self.global_crops_scale,
                             blur_prob=1.0)
        g2 = self._one_group(imgs, self.global_crops_size, self.global_crops_scale,
                             blur_prob=0.1, solarize_prob=0.2)
        locals_ = [
            self._one_group(imgs, self.local_crops_size, self.local_crops_scale,
                            blur_prob=0.5)
            for _ in range(self.local_crops_number)
        ]
        return {
            "global_crops": torch.cat([g1, g2], dim=0),
            "local_crops": torch.cat(locals_, dim=0) if locals_ else
                           imgs.new_zeros(0, 3, self.local_crops_size, self.local_crops_size),
        }
