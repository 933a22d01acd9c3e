"""Collate: crop-major stacking + iBOT mask batch construction.

Parity: dinov3_jax/data/collate.py:16-139, torch-native: output stays a dict
of torch CPU tensors; the trainer moves them to the GPU with pinned buffers +
non-blocking copies (no dlpack bridge, no NCHW->NHWC movedim — our patch-embed
consumes NCHW directly).
"""

from __future__ import annotations

import random
from typing import Optional

import torch


def build_mask_batch(
    B: int,
    n_tokens: int,
    mask_generator,
    mask_ratio_tuple,
    mask_probability: float,
    random_circular_shift: bool = False,
) -> dict:
    """iBOT mask construction for a crop-major batch of B global-crop rows:
    linspace'd mask ratios over mask_probability*B rows, shuffled, with the
    derived mask_indices_list / masks_weight / upperbound (reference
    collate.py:46-80). Host-side (the block-mask generator is sequential)."""
    N = n_tokens
    n_samples_masked = int(B * mask_probability)
    probs = torch.linspace(*mask_ratio_tuple, n_samples_masked + 1)
    upperbound = 0
    masks_list = []
    for i in range(n_samples_masked):
        prob_max = probs[i + 1]
        mask = torch.from_numpy(mask_generator(int(N * prob_max)))
        if random_circular_shift:
            shift_x = random.randint(0, mask.shape[0] - 1)
            shift_y = random.randint(0, mask.shape[1] - 1)
            mask = torch.roll(mask, (shift_x, shift_y), (0, 1))
        masks_list.append(mask)
        upperbound += int(N * prob_max)
    for _ in range(n_samples_masked, B):
        masks_list.append(torch.from_numpy(mask_generator(0)))
    random.shuffle(masks_list)

    collated_masks = torch.stack(masks_list).flatten(1)
    mask_indices_list = collated_masks.flatten().nonzero().flatten()
    masks_weight = (
        (1 / collated_masks.sum(-1).clamp(min=1.0)).unsqueeze(-1).expand_as(collated_masks)[collated_masks]
    )
    return {
        "collated_masks": collated_masks,
        "mask_indices_list": mask_indices_list,
        "masks_weight": masks_weight,
        "upperbound": upperbound,
        "n_masked_patches": torch.full((1,), fill_value=mask_indices_list.shape[0], dtype=torch.long),
    }


def collate_data_and_cast(
    samples_list,
    mask_ratio_tuple,
    mask_probability,
    dtype,
    n_tokens: Optional[int] = None,
    mask_generator=None,
    random_circular_shift: bool = False,
    local_batch_size: Optional[int] = None,
):
    first = samples_list[0][0]
    n_global_crops = len(first["global_crops"])
    n_local_crops = len(first["local_crops"])

    # crop-major layout: [crop0 of all samples, crop1 of all samples, ...]
    collated_global_crops = torch.stack(
        [s[0]["global_crops"][i] for i in range(n_global_crops) for s in samples_list]
    )
    collated_local_crops = torch.stack(
        [s[0]["local_crops"][i] for i in range(n_local_crops) for s in samples_list]
    )
    collated_gram_teacher_crops = None
    if "gram_teacher_crops" in first:
        collated_gram_teacher_crops = torch.stack(
            [s[0]["gram_teacher_crops"][i] for i in range(n_global_crops) for s in samples_list]
        )

    B = n_global_crops * local_batch_size if local_batch_size is not None else len(collated_global_crops)
    mask_batch = build_mask_batch(
        B, n_tokens, mask_generator, mask_ratio_tuple, mask_probability,
        random_circular_shift=random_circular_shift,
    )

    out = {
        "collated_global_crops": collated_global_crops.to(dtype),
        "collated_local_crops": collated_local_crops.to(dtype),
        **mask_batch,
    }
    if collated_gram_teacher_crops is not None:
        out["collated_gram_teacher_crops"] = collated_gram_teacher_crops.to(dtype)
    return out


def get_batch_subset(collated_data_batch: dict, divide_by: int) -> dict:
    """Shrink a collated batch (OOM retry helper; parity collate.py:97-139)."""
    old_bs = collated_data_batch["collated_global_crops"].shape[0] // 2
    target_bs = (old_bs + divide_by - 1) // divide_by
    collated_global_crops = (
        collated_data_batch["collated_global_crops"].unflatten(0, (2, old_bs)).narrow(1, 0, target_bs).flatten(0, 1)
    )
    collated_local_crops = (
        collated_data_batch["collated_local_crops"].unflatten(0, (-1, old_bs)).narrow(1, 0, target_bs).flatten(0, 1)
    )
    masks_old_bs = collated_data_batch["collated_masks"].shape[0] // 2
    masks_target_bs = masks_old_bs // divide_by
    collated_masks = (
        collated_data_batch["collated_masks"].unflatten(0, (2, masks_old_bs)).narrow(1, 0, masks_target_bs).flatten(0, 1)
    )
    mask_indices_list = collated_masks.flatten().nonzero().flatten()
    while mask_indices_list.shape[0] == 0:
        rows = list(collated_data_batch["collated_masks"].unbind(0))
        random.shuffle(rows)
        collated_masks = (
            torch.stack(rows, dim=0).unflatten(0, (2, masks_old_bs)).narrow(1, 0, masks_target_bs).flatten(0, 1)
        )
        mask_indices_list = collated_masks.flatten().nonzero().flatten()
    masks_weight = (
        (1 / collated_masks.sum(-1).clamp(min=1.0)).unsqueeze(-1).expand_as(collated_masks)[collated_masks]
    )
    new_batch = {
        "collated_global_crops": collated_global_crops,
        "collated_local_crops": collated_local_crops,
        "collated_masks": collated_masks,
        "mask_indices_list": mask_indices_list,
        "masks_weight": masks_weight,
        "upperbound": collated_data_batch["upperbound"],
        "n_masked_patches": torch.full((1,), fill_value=mask_indices_list.shape[0], dtype=torch.long),
    }
    if "global_batch_size" in collated_data_batch:
        new_batch["global_batch_size"] = collated_data_batch["global_batch_size"] // divide_by
    return new_batch
