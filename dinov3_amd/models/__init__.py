"""Model factory (parity: dinov3_jax/models/__init__.py:17-98)."""

from __future__ import annotations

import logging

from . import vision_transformer as vits
from .vision_transformer import DinoVisionTransformer  # noqa: F401
from .convnext import ConvNeXt  # noqa: F401

logger = logging.getLogger("dinov3")


def build_model(args, only_teacher: bool = False, img_size: int = 224):
    if "vit" in args.arch:
        vit_kwargs = dict(
            img_size=img_size,
            patch_size=args.patch_size,
            pos_embed_rope_base=args.pos_embed_rope_base,
            pos_embed_rope_min_period=args.pos_embed_rope_min_period,
            pos_embed_rope_max_period=args.pos_embed_rope_max_period,
            pos_embed_rope_normalize_coords=args.pos_embed_rope_normalize_coords,
            pos_embed_rope_shift_coords=args.pos_embed_rope_shift_coords,
            pos_embed_rope_jitter_coords=args.pos_embed_rope_jitter_coords,
            pos_embed_rope_rescale_coords=args.pos_embed_rope_rescale_coords,
            qkv_bias=args.qkv_bias,
            layerscale_init=args.layerscale,
            norm_layer=args.norm_layer,
            ffn_layer=args.ffn_layer,
            ffn_bias=args.ffn_bias,
            proj_bias=args.proj_bias,
            n_storage_tokens=args.n_storage_tokens,
            mask_k_bias=args.mask_k_bias,
            untie_cls_and_patch_norms=args.untie_cls_and_patch_norms,
            untie_global_and_local_cls_norm=args.untie_global_and_local_cls_norm,
        )
        teacher = vits.__dict__[args.arch](**vit_kwargs)
        if only_teacher:
            return teacher, teacher.embed_dim
        student = vits.__dict__[args.arch](**vit_kwargs, drop_path_rate=args.drop_path_rate)
        return student, teacher, student.embed_dim
    if "convnext" in args.arch:
        from . import convnext as cnx

        teacher = cnx.__dict__[args.arch]()
        if only_teacher:
            return teacher, teacher.embed_dim
        student = cnx.__dict__[args.arch](drop_path_rate=args.drop_path_rate)
        return student, teacher, student.embed_dim
    raise NotImplementedError(f"unrecognized architecture {args.arch}")


def build_model_from_cfg(config, only_teacher: bool = False):
    img_size = config.crops.global_crops_size
    if not isinstance(img_size, int):
        img_size = max(img_size)
    return build_model(config.student, only_teacher=only_teacher, img_size=img_size)


def build_model_for_eval(config, pretrained_weights: str = ""):
    model, _ = build_model_from_cfg(config, only_teacher=True)
    if pretrained_weights:
        import torch

        state = torch.load(pretrained_weights, map_location="cpu", weights_only=True)
        if "teacher" in state:
            state = state["teacher"]
        model.load_state_dict(state, strict=False)
        logger.info("loaded eval weights from %s", pretrained_weights)
    model.eval()
    return model
