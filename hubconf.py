"""Weight interop: Meta's PyTorch dinov3 checkpoints <-> dinov3_amd models.

Parity with the reference's hubconf.py (torch->flax converter, hubconf.py:40-80),
in the torch->torch direction our framework needs: the only layout difference
from Meta's ViT is the patch embedding (we store the stride-p conv as a Linear
over the flattened (ph, pw, c) patch so it lowers to one GEMM — SURVEY K1).

Entrypoints follow torch.hub conventions: dinov3_vits16(), dinov3_vitb16(),
dinov3_vitl16(), dinov3_vitg14(), dinov3_vit7b16().
"""

import os
import sys

import torch

REPO_ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO_ROOT)

dependencies = ["torch"]


def convert_meta_state_dict(sd: dict, patch_size: int = 16, in_chans: int = 3) -> dict:
    """Map a Meta dinov3 ViT state_dict onto dinov3_amd.DinoVisionTransformer."""
    out = {}
    for key, value in sd.items():
        if key.endswith("bias_mask"):  # mask_k_bias NaN-buffer: ours is derived
            continue
        if key == "patch_embed.proj.weight" and value.ndim == 4:
            # conv [D, C, ph, pw] -> linear [D, C*ph*pw]: plain reshape (our
            # patch rows use the conv-native (c, dy, dx) flattening)
            value = value.reshape(value.shape[0], -1)
        if key.endswith("last_layer.weight_v"):
            # collapse weight-norm parametrization if present
            g = sd.get(key.replace("weight_v", "weight_g"))
            if g is not None:
                value = value * (g / value.norm(dim=1, keepdim=True))
            key = key.replace("weight_v", "weight")
        if key.endswith("last_layer.weight_g"):
            continue
        if "rope_embed.periods" in key:
            continue  # non-persistent buffer, recomputed from config
        out[key] = value
    return out


def _build(arch: str, weights: str = "", **kwargs):
    from dinov3_amd.models import vision_transformer as vits

    model = vits.__dict__[arch](**kwargs)
    if weights:
        sd = torch.load(weights, map_location="cpu", weights_only=True)
        if "teacher" in sd:
            sd = sd["teacher"]
        sd = {k.replace("backbone.", ""): v for k, v in sd.items()}
        missing, unexpected = model.load_state_dict(convert_meta_state_dict(sd), strict=False)
        if missing or unexpected:
            print(f"hubconf: partial load (missing {len(missing)}, unexpected {len(unexpected)})")
    model.eval()
    return model


def dinov3_vits16(weights: str = "", **kwargs):
    return _build("vit_small", weights, layerscale_init=1e-5, **kwargs)


def dinov3_vitb16(weights: str = "", **kwargs):
    return _build("vit_base", weights, layerscale_init=1e-5, **kwargs)


def dinov3_vitl16(weights: str = "", **kwargs):
    return _build("vit_large", weights, layerscale_init=1e-5, **kwargs)


def dinov3_vitg14(weights: str = "", **kwargs):
    return _build("vit_giant2", weights, patch_size=14, layerscale_init=1e-5,
                  ffn_layer="swiglu64", **kwargs)


def dinov3_vit7b16(weights: str = "", **kwargs):
    return _build("vit_7b", weights, layerscale_init=1e-5, ffn_layer="swiglu64",
                  n_storage_tokens=4, mask_k_bias=True, norm_layer="layernormbf16",
                  untie_cls_and_patch_norms=True, **kwargs)
