from .attention import LinearKMaskedBias, SelfAttention
from .causal_attention import CausalSelfAttention, CausalSelfAttentionBlock
from .block import SelfAttentionBlock
from .dino_head import DINOHead
from .ffn_layers import FFN_LAYERS, Mlp, SwiGLUFFN
from .norms import NORM_LAYERS, LayerNorm, LayerScale, RMSNorm
from .patch_embed import PatchEmbed
from .rope import RopePositionEmbedding

__all__ = [
    "SelfAttention",
    "CausalSelfAttention",
    "CausalSelfAttentionBlock",
    "LinearKMaskedBias",
    "SelfAttentionBlock",
    "DINOHead",
    "Mlp",
    "SwiGLUFFN",
    "FFN_LAYERS",
    "LayerNorm",
    "RMSNorm",
    "LayerScale",
    "NORM_LAYERS",
    "PatchEmbed",
    "RopePositionEmbedding",
]
