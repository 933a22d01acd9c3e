"""MLP forward/backward on hipBLASLt fused epilogues — PROBED AND REJECTED.

Kept as evidence/experiment scaffolding only. On this stack (ROCm 7.2
hipBLASLt, gfx950) the heuristic returns ZERO algorithms for every aux /
gradient epilogue (GELU_AUX, GELU_AUX_BIAS, DGELU, DGELU_BGRAD, BGRADA,
BGRADB) — only BIAS and GELU_BIAS exist (probed on hardware with
blaslt_probe_epilogue, 2026-09; 8 algorithms each for BIAS/GELU_BIAS, 0 for
all others). Without GELU_AUX the fused forward cannot save the
pre-activation that backward needs, so GEMM-epilogue MLP fusion is not
viable here and the hand-written bias_gelu kernels (ops/csrc/elementwise.hip)
remain the production path. This module raises if the entry points hit an
unsupported epilogue.

Reference behavior: dinov3_jax/layers/ffn_layers.py:24-49 (minus the §8 B4
extra activation after fc2, deliberately not reproduced).
"""

from __future__ import annotations

import os

import torch

__all__ = ["blaslt_mlp_enabled", "blaslt_mlp"]


def blaslt_mlp_enabled(x: torch.Tensor) -> bool:
    from . import use_hip

    return (os.environ.get("DINOV3_BLASLT_MLP", "0") == "1" and use_hip(x)
            and x.dtype == torch.bfloat16)


class _BlasLtMlpFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        from . import hip_ops

        x2d = x.reshape(-1, x.shape[-1])
        h, pre = hip_ops().blaslt_gemm_bias_gelu_fwd(x2d, w1, b1)
        y = torch.nn.functional.linear(h, w2, b2)
        ctx.save_for_backward(x2d, w1, w2, pre, h)
        ctx.in_shape = x.shape
        return y.reshape(*x.shape[:-1], w2.shape[0])

    @staticmethod
    def backward(ctx, dy):
        from . import hip_ops

        x2d, w1, w2, pre, h = ctx.saved_tensors
        dy2d = dy.reshape(-1, dy.shape[-1]).contiguous()
        # fused: dpre = gelu'(pre) * (dy @ w2); db1 = colsum(dpre)
        dpre, db1 = hip_ops().blaslt_gemm_dgelu_bgrad(dy2d, w2, pre)
        dw2 = dy2d.transpose(0, 1) @ h
        db2 = dy2d.sum(dim=0)
        dw1 = dpre.transpose(0, 1) @ x2d
        dx = dpre @ w1
        return dx.reshape(ctx.in_shape), dw1, db1.to(w1.dtype), dw2, db2


def blaslt_mlp(x: torch.Tensor, w1: torch.Tensor, b1: torch.Tensor,
               w2: torch.Tensor, b2: torch.Tensor) -> torch.Tensor:
    """y = gelu_tanh(x@W1^T + b1) @ W2^T + b2 with epilogue-fused GEMMs."""
    return _BlasLtMlpFn.apply(x, w1, b1, w2, b2)
