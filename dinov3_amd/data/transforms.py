"""Torch-native image transform primitives (no torchvision/PIL dependency).

Images are float32 CHW tensors in [0,1] (datasets decode uint8 HWC -> this).
Covers what DataAugmentationDINO needs (reference dinov3_jax/data/
augmentations.py + transforms.py): random resized crop, hflip, color jitter,
grayscale, gaussian blur, solarize, normalize, plus eval presets.
"""

from __future__ import annotations

import math
import random
from typing import Optional, Sequence, Tuple

import torch
import torch.nn.functional as F


def resize(img: torch.Tensor, size: int, mode: str = "bicubic") -> torch.Tensor:
    return F.interpolate(img.unsqueeze(0), size=(size, size), mode=mode, align_corners=False,
                         antialias=True).squeeze(0).clamp(0.0, 1.0)


def center_crop(img: torch.Tensor, size: int) -> torch.Tensor:
    _, h, w = img.shape
    top = max((h - size) // 2, 0)
    left = max((w - size) // 2, 0)
    return img[:, top: top + size, left: left + size]


def sample_rrc_box(h: int, w: int, scale: Tuple[float, float],
                   ratio: Tuple[float, float] = (3.0 / 4.0, 4.0 / 3.0)) -> Tuple[int, int, int, int]:
    area = h * w
    for _ in range(10):
        target_area = area * random.uniform(*scale)
        log_ratio = (math.log(ratio[0]), math.log(ratio[1]))
        aspect = math.exp(random.uniform(*log_ratio))
        cw = int(round(math.sqrt(target_area * aspect)))
        ch = int(round(math.sqrt(target_area / aspect)))
        if 0 < cw <= w and 0 < ch <= h:
            top = random.randint(0, h - ch)
            left = random.randint(0, w - cw)
            return top, left, ch, cw
    # fallback: central crop at clamped aspect
    in_ratio = w / h
    if in_ratio < ratio[0]:
        cw, ch = w, int(round(w / ratio[0]))
    elif in_ratio > ratio[1]:
        ch, cw = h, int(round(h * ratio[1]))
    else:
        cw, ch = w, h
    return (h - ch) // 2, (w - cw) // 2, ch, cw


def random_resized_crop(img: torch.Tensor, size: int, scale: Tuple[float, float],
                        box: Optional[Tuple[int, int, int, int]] = None) -> torch.Tensor:
    _, h, w = img.shape
    if box is None:
        box = sample_rrc_box(h, w, scale)
    top, left, ch, cw = box
    crop = img[:, top: top + ch, left: left + cw]
    return resize(crop, size)


def hflip(img: torch.Tensor) -> torch.Tensor:
    return img.flip(-1)


def _blend(a: torch.Tensor, b: torch.Tensor, alpha: float) -> torch.Tensor:
    return (alpha * a + (1.0 - alpha) * b).clamp(0.0, 1.0)


def rgb_to_grayscale(img: torch.Tensor) -> torch.Tensor:
    g = (0.2989 * img[0] + 0.587 * img[1] + 0.114 * img[2]).unsqueeze(0)
    return g.expand_as(img).contiguous()


def adjust_brightness(img: torch.Tensor, factor: float) -> torch.Tensor:
    return _blend(img, torch.zeros_like(img), factor)


def adjust_contrast(img: torch.Tensor, factor: float) -> torch.Tensor:
    mean = rgb_to_grayscale(img).mean()
    return _blend(img, mean.expand_as(img), factor)


def adjust_saturation(img: torch.Tensor, factor: float) -> torch.Tensor:
    return _blend(img, rgb_to_grayscale(img), factor)


def adjust_hue(img: torch.Tensor, hue_shift: float) -> torch.Tensor:
    """hue_shift in [-0.5, 0.5] turns of the hue wheel (branch-free HSV trip)."""
    r, g, b = img[0], img[1], img[2]
    maxc, _ = img.max(dim=0)
    minc, _ = img.min(dim=0)
    v = maxc
    deltac = maxc - minc
    s = deltac / maxc.clamp_min(1e-8)
    dz = deltac.clamp_min(1e-8)
    rc = (maxc - r) / dz
    gc = (maxc - g) / dz
    bc = (maxc - b) / dz
    h = torch.where(r == maxc, bc - gc, torch.where(g == maxc, 2.0 + rc - bc, 4.0 + gc - rc))
    h6 = (h % 6.0) + hue_shift * 6.0
    # hsv -> rgb without per-sector masks: c(n) = v - v*s*clamp(min(k, 4-k), 0, 1)
    out = torch.empty_like(img)
    for ch, n in enumerate((5.0, 3.0, 1.0)):
        k = (n + h6) % 6.0
        out[ch] = v - v * s * torch.clamp(torch.minimum(k, 4.0 - k), 0.0, 1.0)
    return out.clamp(0.0, 1.0)


def color_jitter(img: torch.Tensor, brightness: float, contrast: float,
                 saturation: float, hue: float) -> torch.Tensor:
    ops = []
    if brightness > 0:
        ops.append(("b", random.uniform(max(0.0, 1 - brightness), 1 + brightness)))
    if contrast > 0:
        ops.append(("c", random.uniform(max(0.0, 1 - contrast), 1 + contrast)))
    if saturation > 0:
        ops.append(("s", random.uniform(max(0.0, 1 - saturation), 1 + saturation)))
    if hue > 0:
        ops.append(("h", random.uniform(-hue, hue)))
    random.shuffle(ops)
    for kind, val in ops:
        if kind == "b":
            img = adjust_brightness(img, val)
        elif kind == "c":
            img = adjust_contrast(img, val)
        elif kind == "s":
            img = adjust_saturation(img, val)
        else:
            img = adjust_hue(img, val)
    return img


def gaussian_blur(img: torch.Tensor, sigma: float) -> torch.Tensor:
    ksize = max(int(2 * round(3.0 * sigma) + 1), 3)
    x = torch.arange(ksize, dtype=torch.float32) - ksize // 2
    kernel = torch.exp(-0.5 * (x / sigma) ** 2)
    kernel = kernel / kernel.sum()
    c = img.shape[0]
    kx = kernel.view(1, 1, 1, ksize).expand(c, 1, 1, ksize)
    ky = kernel.view(1, 1, ksize, 1).expand(c, 1, ksize, 1)
    pad = ksize // 2
    out = F.conv2d(img.unsqueeze(0), kx, padding=(0, pad), groups=c)
    out = F.conv2d(out, ky, padding=(pad, 0), groups=c)
    return out.squeeze(0)


def solarize(img: torch.Tensor, threshold: float = 0.5) -> torch.Tensor:
    return torch.where(img >= threshold, 1.0 - img, img)


def normalize(img: torch.Tensor, mean: Sequence[float], std: Sequence[float]) -> torch.Tensor:
    mean_t = torch.tensor(mean, dtype=img.dtype).view(-1, 1, 1)
    std_t = torch.tensor(std, dtype=img.dtype).view(-1, 1, 1)
    return (img - mean_t) / std_t


def make_eval_transform(resize_size: int = 256, crop_size: int = 224,
                        mean=(0.485, 0.456, 0.406), std=(0.229, 0.224, 0.225),
                        resize_square: bool = False, resize_large_side: bool = False,
                        mode: str = "bicubic"):
    """Eval preset (reference transforms.py:106-131): resize (short side,
    square, or long side) -> optional center crop -> normalize."""
    assert not (resize_square and resize_large_side)

    def _t(img: torch.Tensor) -> torch.Tensor:
        if resize_square:
            img = torch.nn.functional.interpolate(
                img.unsqueeze(0), size=(resize_size, resize_size), mode=mode,
                align_corners=False, antialias=True).squeeze(0).clamp(0, 1)
        elif resize_large_side:
            c, h, w = img.shape
            s = resize_size / max(h, w)
            nh, nw = max(1, round(h * s)), max(1, round(w * s))
            img = torch.nn.functional.interpolate(
                img.unsqueeze(0), size=(nh, nw), mode=mode,
                align_corners=False, antialias=True).squeeze(0).clamp(0, 1)
        else:
            img = resize(img, resize_size, mode)
        if crop_size:
            img = center_crop(img, crop_size)
        return normalize(img, mean, std)

    return _t


def make_classification_eval_transform(resize_size: int = 256, crop_size: int = 224,
                                       mean=(0.485, 0.456, 0.406),
                                       std=(0.229, 0.224, 0.225)):
    """Torchvision-style classification eval preset (reference
    transforms.py:134-150)."""
    return make_eval_transform(resize_size, crop_size, mean, std)


def make_classification_train_transform(crop_size: int = 224, hflip_prob: float = 0.5,
                                        mean=(0.485, 0.456, 0.406),
                                        std=(0.229, 0.224, 0.225)):
    """Torchvision-style classification train preset (reference
    transforms.py:66-80): RandomResizedCrop -> flip -> normalize."""
    import random as _random

    def _t(img: torch.Tensor) -> torch.Tensor:
        img = random_resized_crop(img, crop_size, (0.08, 1.0))
        if hflip_prob > 0 and _random.random() < hflip_prob:
            img = hflip(img)
        return normalize(img, mean, std)

    return _t


def voc2007_classification_target_transform(label, n_categories: int = 20) -> torch.Tensor:
    """Multi-label one-hot from a VOC-style label with .instances
    (reference transforms.py:153-157)."""
    one_hot = torch.zeros(n_categories, dtype=torch.long)
    for instance in label.instances:
        one_hot[instance.category_id] = 1
    return one_hot


def imaterialist_classification_target_transform(label, n_categories: int = 294) -> torch.Tensor:
    one_hot = torch.zeros(n_categories, dtype=torch.long)
    one_hot[label.attributes] = 1
    return one_hot


def get_target_transform(dataset_str: str):
    """Dataset-string -> target transform (reference transforms.py:166-170)."""
    if "VOC2007" in dataset_str:
        return voc2007_classification_target_transform
    if "IMaterialist" in dataset_str:
        return imaterialist_classification_target_transform
    return None
