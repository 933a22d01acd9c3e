"""FFN layers: Mlp (fc1 -> fused bias+GELU -> fc2) and SwiGLUFFN.

Reference: dinov3_jax/layers/ffn_layers.py. The reference Mlp applies an extra
activation+dropout after fc2 (bug) — standard ViT semantics here.
SwiGLU hidden sizing/alignment map: vision_transformer.py:31-37.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import bias_gelu
from ..ops.bias_act import swiglu_gate


class Mlp(nn.Module):
    def __init__(self, in_features: int, hidden_features: int, out_features: int = None,
                 drop: float = 0.0, use_bias: bool = True, **kwargs):
        super().__init__()
        out_features = out_features or in_features
        self.fc1 = nn.Linear(in_features, hidden_features, bias=use_bias)
        self.fc2 = nn.Linear(hidden_features, out_features, bias=use_bias)
        self.drop = nn.Dropout(drop) if drop > 0 else nn.Identity()

    def forward(self, x: torch.Tensor, skip_out_bias: bool = False) -> torch.Tensor:
        # fc1 GEMM without bias epilogue; bias fused into the GELU kernel.
        # (hipBLASLt epilogue fusion was probed on gfx950 and rejected: only
        # BIAS/GELU_BIAS exist — no GELU_AUX / DGELU_BGRAD — so a GEMM-fused
        # forward would leave backward without the pre-activation. See
        # ops/csrc/blaslt_ext.hip and docs/KERNELS.md.)
        # skip_out_bias: caller folds fc2's bias into the residual kernel.
        h = F.linear(x, self.fc1.weight)
        h = bias_gelu(h, self.fc1.bias)
        h = self.drop(h)
        if skip_out_bias:
            return self.drop(F.linear(h, self.fc2.weight))
        return self.drop(self.fc2(h))


class SwiGLUFFN(nn.Module):
    def __init__(self, in_features: int, hidden_features: int, out_features: int = None,
                 drop: float = 0.0, use_bias: bool = True, align_to: int = 8, **kwargs):
        super().__init__()
        out_features = out_features or in_features
        # hidden = 2/3 * h rounded up to the alignment (reference
        # vision_transformer.py:31-37 swiglu/swiglu32/64/128 variants).
        d = int(hidden_features * 2 / 3)
        swiglu_hidden = d + (-d) % align_to
        self.hidden_features = swiglu_hidden
        self.w12 = nn.Linear(in_features, 2 * swiglu_hidden, bias=use_bias)
        self.w3 = nn.Linear(swiglu_hidden, out_features, bias=use_bias)
        self.drop = nn.Dropout(drop) if drop > 0 else nn.Identity()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x12 = self.w12(x)
        h = swiglu_gate(x12)
        return self.drop(self.w3(h))


FFN_LAYERS = {
    "mlp": Mlp,
    "swiglu": lambda **kw: SwiGLUFFN(align_to=8, **kw),
    "swiglu32": lambda **kw: SwiGLUFFN(align_to=32, **kw),
    "swiglu64": lambda **kw: SwiGLUFFN(align_to=64, **kw),
    "swiglu128": lambda **kw: SwiGLUFFN(align_to=128, **kw),
}
