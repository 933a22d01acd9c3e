"""Per-parameter lr/wd multipliers + fusing into multi-tensor groups.

Parity: dinov3_jax/train/param_groups.py:56-160. Rules:
- layerwise lr decay: lr_decay_rate ** (n_blocks + 1 - layer_id), layer_id 0
  for patch_embed/tokens, block_index+1 for blocks, n_blocks+1 otherwise;
- patch_embed lr multiplier on top;
- wd multiplier 0 for biases / norm params / layerscale gamma;
- dino_head wd multiplier;
- is_last_layer flags the prototype layer (lr frozen for the first epochs).

Params with identical (lr_mult, wd_mult, is_last_layer) fuse into one group —
one multi-tensor AdamW kernel launch per group per step.
"""

from __future__ import annotations

import logging
import re
from collections import defaultdict
from typing import Dict, List, Tuple

import torch

logger = logging.getLogger("dinov3")


def get_vit_lr_decay_rate(name: str, lr_decay_rate: float = 1.0, num_layers: int = 12) -> float:
    layer_id = num_layers + 1
    if any(tok in name for tok in ("pos_embed", "patch_embed", "mask_token", "cls_token", "storage_tokens")):
        layer_id = 0
    else:
        m = re.search(r"blocks\.(\d+)\.", name)
        if m is not None and "residual" not in name:
            layer_id = int(m.group(1)) + 1
    return lr_decay_rate ** (num_layers + 1 - layer_id)


def get_params_groups_with_decay(
    submodels: Dict[str, torch.nn.Module],
    lr_decay_rate: float = 1.0,
    patch_embed_lr_mult: float = 1.0,
    dino_head_wd_multiplier: float = 1.0,
) -> List[dict]:
    """submodels: {"backbone": module, "dino_head": module, "ibot_head": module}.

    Returns a list of fused group dicts:
    {params, names, lr_multiplier, wd_multiplier, is_last_layer}.
    """
    buckets: Dict[Tuple[str, float, float, bool], dict] = defaultdict(
        lambda: {"params": [], "names": []}
    )
    for root_name, model in submodels.items():
        n_blocks = len(getattr(model, "blocks", [])) if hasattr(model, "blocks") else 0
        for name, param in model.named_parameters():
            if not param.requires_grad:
                continue
            full_name = f"{root_name}.{name}"
            lr_mult = get_vit_lr_decay_rate(name, lr_decay_rate, num_layers=n_blocks) if n_blocks > 0 else 1.0
            wd_mult = 1.0
            is_last_layer = "last_layer" in name
            if "dino_head" in root_name:
                wd_mult = dino_head_wd_multiplier
            if name.endswith("bias") or "norm" in name or "gamma" in name:
                wd_mult = 0.0
            if "patch_embed" in name:
                lr_mult *= patch_embed_lr_mult
            # groups fuse within a submodel only (grad clipping is per-submodel)
            key = (root_name, lr_mult, wd_mult, is_last_layer)
            buckets[key]["params"].append(param)
            buckets[key]["names"].append(full_name)
    groups = []
    for (root_name, lr_mult, wd_mult, is_last_layer), b in sorted(buckets.items(), key=lambda kv: kv[0]):
        groups.append(
            {
                "params": b["params"],
                "names": b["names"],
                "submodel": root_name,
                "lr_multiplier": lr_mult,
                "wd_multiplier": wd_mult,
                "is_last_layer": is_last_layer,
            }
        )
    logger.info("fused %d param groups", len(groups))
    return groups
