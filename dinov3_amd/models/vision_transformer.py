"""DINOv3 Vision Transformer, MI355X-native.

Capability parity with dinov3_jax/models/vision_transformer.py:56-408:
patch embed -> [cls | storage | patches(+mask-token substitution)] -> N x
pre-norm blocks with per-resolution RoPE -> final norm with optional untied
cls/patch and local/global-cls norms. Size ctors vit_small .. vit_7b.

Differences by design: the multi-crop list-forward batches the global+local
crop groups through shared GEMMs (see layers/block.py), dtype is managed by
the trainer (bf16 params / fp32 reference on CPU), and sharding is an external
engine (parallel/fsdp.py) instead of a module wrapper.
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Sequence, Tuple, Union

import torch
import torch.nn as nn

from ..layers import (
    FFN_LAYERS,
    NORM_LAYERS,
    PatchEmbed,
    RopePositionEmbedding,
    SelfAttentionBlock,
)

logger = logging.getLogger("dinov3")

# DINOV3_NAN_CHECK=1: per-block non-finite activation sanitizer (debug mode;
# the SURVEY §5 compute-sanitizer analogue — costs a device sync per block)
import os as _os

_NAN_CHECK = _os.environ.get("DINOV3_NAN_CHECK", "0") == "1"


class DinoVisionTransformer(nn.Module):
    def __init__(
        self,
        img_size: int = 224,
        patch_size: int = 16,
        in_chans: int = 3,
        pos_embed_rope_base: Optional[float] = 100.0,
        pos_embed_rope_min_period: Optional[float] = None,
        pos_embed_rope_max_period: Optional[float] = None,
        pos_embed_rope_normalize_coords: str = "separate",
        pos_embed_rope_shift_coords: Optional[float] = None,
        pos_embed_rope_jitter_coords: Optional[float] = None,
        pos_embed_rope_rescale_coords: Optional[float] = None,
        pos_embed_rope_dtype: str = "bf16",
        embed_dim: int = 768,
        n_blocks: int = 12,
        num_heads: int = 12,
        ffn_ratio: float = 4.0,
        qkv_bias: bool = True,
        drop_path_rate: float = 0.0,
        layerscale_init: Optional[float] = None,
        norm_layer: str = "layernorm",
        ffn_layer: str = "mlp",
        ffn_bias: bool = True,
        proj_bias: bool = True,
        n_storage_tokens: int = 0,
        mask_k_bias: bool = False,
        untie_cls_and_patch_norms: bool = False,
        untie_global_and_local_cls_norm: bool = False,
    ):
        super().__init__()
        self.img_size = img_size
        self.patch_size = patch_size
        self.embed_dim = embed_dim
        self.num_features = embed_dim
        self.n_blocks = n_blocks
        self.num_heads = num_heads
        self.n_storage_tokens = n_storage_tokens
        self.untie_cls_and_patch_norms = untie_cls_and_patch_norms
        self.untie_global_and_local_cls_norm = untie_global_and_local_cls_norm

        norm_factory = NORM_LAYERS[norm_layer]
        ffn_factory = FFN_LAYERS[ffn_layer]

        self.patch_embed = PatchEmbed(
            img_size=img_size, patch_size=patch_size, in_chans=in_chans, embed_dim=embed_dim,
        )
        self.cls_token = nn.Parameter(torch.empty(1, 1, embed_dim).normal_(std=0.02))
        if n_storage_tokens > 0:
            self.storage_tokens = nn.Parameter(torch.empty(1, n_storage_tokens, embed_dim).normal_(std=0.02))
        else:
            self.storage_tokens = None
        self.mask_token = nn.Parameter(torch.zeros(1, embed_dim))

        self.rope_embed = RopePositionEmbedding(
            embed_dim=embed_dim,
            num_heads=num_heads,
            base=pos_embed_rope_base,
            min_period=pos_embed_rope_min_period,
            max_period=pos_embed_rope_max_period,
            normalize_coords=pos_embed_rope_normalize_coords,
            shift_coords=pos_embed_rope_shift_coords,
            jitter_coords=pos_embed_rope_jitter_coords,
            rescale_coords=pos_embed_rope_rescale_coords,
        )

        self.blocks = nn.ModuleList(
            SelfAttentionBlock(
                dim=embed_dim,
                num_heads=num_heads,
                ffn_ratio=ffn_ratio,
                qkv_bias=qkv_bias,
                proj_bias=proj_bias,
                ffn_bias=ffn_bias,
                drop_path=drop_path_rate,
                norm_layer=norm_factory,
                ffn_layer=ffn_factory,
                init_values=layerscale_init,
                mask_k_bias=mask_k_bias,
            )
            for _ in range(n_blocks)
        )
        self.norm = norm_factory(embed_dim)
        self.cls_norm = norm_factory(embed_dim) if untie_cls_and_patch_norms else None
        self.local_cls_norm = norm_factory(embed_dim) if untie_global_and_local_cls_norm else None
        self.grad_checkpointing = False  # selective recompute per block (288 GB
        # HBM makes this OFF by default up to ViT-g; flip for 7B-class runs)
        self._init_weights()

    def set_grad_checkpointing(self, enabled: bool = True) -> None:
        self.grad_checkpointing = enabled

    def _init_weights(self) -> None:
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.trunc_normal_(m.weight, std=0.02)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)

    def prepare_tokens_with_masks(self, x: torch.Tensor,
                                  masks: Optional[torch.Tensor] = None) -> Tuple[torch.Tensor, Tuple[int, int]]:
        tokens, hp, wp = self.patch_embed(x)  # [B, N, D]
        B = tokens.shape[0]
        if masks is not None:
            tokens = torch.where(masks.unsqueeze(-1), self.mask_token.to(tokens.dtype).unsqueeze(0), tokens)
        parts = [self.cls_token.to(tokens.dtype).expand(B, -1, -1)]
        if self.storage_tokens is not None:
            parts.append(self.storage_tokens.to(tokens.dtype).expand(B, -1, -1))
        parts.append(tokens)
        return torch.cat(parts, dim=1), (hp, wp)

    def _final_norm(self, x: torch.Tensor, idx: int) -> Tuple[torch.Tensor, torch.Tensor]:
        n_prefix = self.n_storage_tokens + 1
        if self.untie_cls_and_patch_norms or self.untie_global_and_local_cls_norm:
            if self.untie_global_and_local_cls_norm and self.training and idx == 1:
                x_cls_reg = self.local_cls_norm(x[:, :n_prefix])
            elif self.untie_cls_and_patch_norms:
                x_cls_reg = self.cls_norm(x[:, :n_prefix])
            else:
                x_cls_reg = self.norm(x[:, :n_prefix])
            x_patch = self.norm(x[:, n_prefix:])
        else:
            x_norm = self.norm(x)
            x_cls_reg = x_norm[:, :n_prefix]
            x_patch = x_norm[:, n_prefix:]
        return x_cls_reg, x_patch

    def forward_features_list(self, x_list: Sequence[torch.Tensor],
                              masks_list: Sequence[Optional[torch.Tensor]]) -> List[Dict[str, torch.Tensor]]:
        tokens: List[torch.Tensor] = []
        hw: List[Tuple[int, int]] = []
        for t_x, t_masks in zip(x_list, masks_list):
            t, hw_t = self.prepare_tokens_with_masks(t_x, t_masks)
            tokens.append(t)
            hw.append(hw_t)

        # one RoPE table per distinct resolution per step (tables are shared
        # across blocks; the reference recomputes per block — same values)
        rope_tables = {}
        for (H, W) in hw:
            if (H, W) not in rope_tables:
                # tables stay fp32: the HIP rope+fmha kernel applies them in
                # fp32 and writes back bf16 (reference round-trips through
                # bf16 tables; fp32 here is a strict accuracy improvement)
                sin, cos = self.rope_embed(H=H, W=W, training=self.training, device=tokens[0].device)
                rope_tables[(H, W)] = (sin.contiguous(), cos.contiguous())

        # flatten all crop groups into ONE [R, D] buffer for the whole depth
        from ..layers.attention import SelfAttention
        from ..utils.utils import cat_keep_shapes, uncat_with_shapes

        flat, shapes, counts = cat_keep_shapes(tokens)
        metas = []
        off = 0
        for t, key in zip(tokens, hw):
            metas.append(SelfAttention._meta_for(t, rope_tables[key], off))
            off += t.shape[0] * t.shape[1]

        plan = None
        if self.training and self.blocks[0].sample_drop_ratio > 0.0:
            from ..layers.block import DropPathPlan

            keep_ratio = 1.0 - self.blocks[0].sample_drop_ratio
            plan = DropPathPlan(metas, keep_ratio, 2 * len(self.blocks), flat.device)

        if self.grad_checkpointing and self.training and torch.is_grad_enabled():
            from torch.utils.checkpoint import checkpoint

            for i, block in enumerate(self.blocks):
                # inplace_ok=False: the checkpoint saves `flat` for recompute,
                # so the fused residual scatter must not mutate it in place.
                flat = checkpoint(block.forward_flat, flat, metas, plan, i, False,
                                  use_reentrant=False)
        else:
            for i, block in enumerate(self.blocks):
                flat = block.forward_flat(flat, metas, plan, i)
                if _NAN_CHECK and not torch.isfinite(flat).all():
                    raise FloatingPointError(
                        f"non-finite activations after block {i} "
                        f"(DINOV3_NAN_CHECK sanitizer)")
        tokens = uncat_with_shapes(flat, shapes, counts)

        output = []
        for idx, (x, masks) in enumerate(zip(tokens, masks_list)):
            x_cls_reg, x_patch = self._final_norm(x, idx)
            output.append(
                {
                    "x_norm_clstoken": x_cls_reg[:, 0],
                    "x_storage_tokens": x_cls_reg[:, 1:],
                    "x_norm_patchtokens": x_patch,
                    "x_prenorm": x,
                    "masks": masks,
                }
            )
        return output

    def forward_features(self, x: Union[torch.Tensor, Sequence[torch.Tensor]],
                         masks: Union[None, torch.Tensor, Sequence[Optional[torch.Tensor]]] = None):
        if isinstance(x, torch.Tensor):
            return self.forward_features_list([x], [masks])[0]
        if masks is None:
            masks = [None] * len(x)
        return self.forward_features_list(x, masks)

    def _get_intermediate_layers_not_chunked(self, x: torch.Tensor, n: Union[int, Sequence[int]] = 1):
        tokens, (H, W) = self.prepare_tokens_with_masks(x)
        total = len(self.blocks)
        blocks_to_take = range(total - n, total) if isinstance(n, int) else n
        sin, cos = self.rope_embed(H=H, W=W, training=self.training, device=tokens.device)
        rope = (sin, cos)
        output = []
        for i, block in enumerate(self.blocks):
            tokens = block(tokens, rope)
            if i in blocks_to_take:
                output.append(tokens)
        assert len(output) == len(list(blocks_to_take))
        return output

    def get_intermediate_layers(
        self,
        x: torch.Tensor,
        *,
        n: Union[int, Sequence[int]] = 1,
        reshape: bool = False,
        return_class_token: bool = False,
        return_extra_tokens: bool = False,
        norm: bool = True,
    ):
        outputs = self._get_intermediate_layers_not_chunked(x, n)
        n_prefix = self.n_storage_tokens + 1
        if norm:
            normed = []
            for out in outputs:
                if self.untie_cls_and_patch_norms:
                    cls_reg = self.cls_norm(out[:, :n_prefix])
                    patch = self.norm(out[:, n_prefix:])
                    normed.append(torch.cat([cls_reg, patch], dim=1))
                else:
                    normed.append(self.norm(out))
            outputs = normed
        class_tokens = [out[:, 0] for out in outputs]
        extra_tokens = [out[:, 1:n_prefix] for out in outputs]
        patch_outputs = [out[:, n_prefix:] for out in outputs]
        if reshape:
            B, _, h, w = x.shape
            patch_outputs = [
                out.reshape(B, h // self.patch_size, w // self.patch_size, -1).permute(0, 3, 1, 2).contiguous()
                for out in patch_outputs
            ]
        if not return_class_token and not return_extra_tokens:
            return tuple(patch_outputs)
        if return_class_token and not return_extra_tokens:
            return tuple(zip(patch_outputs, class_tokens))
        if not return_class_token and return_extra_tokens:
            return tuple(zip(patch_outputs, extra_tokens))
        return tuple(zip(patch_outputs, class_tokens, extra_tokens))

    def forward(self, x, masks=None, is_training: bool = False):
        ret = self.forward_features(x, masks)
        if is_training:
            return ret
        if isinstance(ret, list):
            return [r["x_norm_clstoken"] for r in ret]
        return ret["x_norm_clstoken"]


def vit_small(patch_size=16, **kwargs):
    return DinoVisionTransformer(patch_size=patch_size, embed_dim=384, n_blocks=12, num_heads=6, ffn_ratio=4, **kwargs)


def vit_base(patch_size=16, **kwargs):
    return DinoVisionTransformer(patch_size=patch_size, embed_dim=768, n_blocks=12, num_heads=12, ffn_ratio=4, **kwargs)


def vit_large(patch_size=16, **kwargs):
    return DinoVisionTransformer(patch_size=patch_size, embed_dim=1024, n_blocks=24, num_heads=16, ffn_ratio=4, **kwargs)


def vit_so400m(patch_size=16, **kwargs):
    return DinoVisionTransformer(
        patch_size=patch_size, embed_dim=1152, n_blocks=27, num_heads=18, ffn_ratio=3.777777778, **kwargs
    )


def vit_huge2(patch_size=16, **kwargs):
    return DinoVisionTransformer(patch_size=patch_size, embed_dim=1280, n_blocks=32, num_heads=20, ffn_ratio=4, **kwargs)


def vit_giant2(patch_size=16, **kwargs):
    return DinoVisionTransformer(patch_size=patch_size, embed_dim=1536, n_blocks=40, num_heads=24, ffn_ratio=4, **kwargs)


def vit_7b(patch_size=16, **kwargs):
    return DinoVisionTransformer(patch_size=patch_size, embed_dim=4096, n_blocks=40, num_heads=32, ffn_ratio=3, **kwargs)
