"""Fused multi-tensor update kernels: EMA teacher (K23), AdamW (K24), grad-norm (K25).

GPU path: one kernel launch walks a packed table of tensor chunks
(csrc/multi_tensor.hip) — the EMA update over all student shards is a single
in-place HIP kernel, as required by the north-star. CPU path: torch._foreach.

Semantics:
- EMA: t = m*t + (1-m)*s over matching (teacher, student) leaves
  (reference ssl_meta_arch.py:650-659, with the wiring bug §8 B2 fixed: the
  LIVE teacher params are updated).
- AdamW: decoupled weight decay; per-group lr/wd multipliers folded in by the
  caller (reference train.py:95-106 via optax; B3 late-binding bug fixed).
"""

from __future__ import annotations

from typing import List, Optional

import torch


def ema_update_(teacher_params: List[torch.Tensor], student_params: List[torch.Tensor], momentum: float) -> None:
    if not teacher_params:
        return
    if teacher_params[0].is_cuda:
        from . import hip_ops

        hip_ops().multi_tensor_ema(teacher_params, student_params, momentum)
        return
    torch._foreach_mul_(teacher_params, momentum)
    torch._foreach_add_(teacher_params, student_params, alpha=1.0 - momentum)


def grad_l2_norm_sq(grads: List[torch.Tensor]) -> torch.Tensor:
    """Sum of squares over a list of grads (fp32 scalar on the same device)."""
    if not grads:
        return torch.zeros((), dtype=torch.float32)
    if grads[0].is_cuda:
        from . import hip_ops

        return hip_ops().multi_tensor_l2norm_sq(grads)
    return sum(g.float().pow(2).sum() for g in grads)


@torch.no_grad()
def multi_tensor_adamw_(
    params: List[torch.Tensor],
    grads: List[torch.Tensor],
    exp_avg: List[torch.Tensor],
    exp_avg_sq: List[torch.Tensor],
    master: Optional[List[torch.Tensor]],
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step: int,
    grad_scale: float = 1.0,
) -> None:
    """AdamW step on one fused group.

    `master` (optional) are fp32 master weights when `params` are bf16; the
    update runs on master and params are re-quantized from it.
    `grad_scale` multiplies grads on the fly (used for clipping).
    """
    if not params:
        return
    bc1 = 1.0 - beta1**step
    bc2 = 1.0 - beta2**step
    if params[0].is_cuda:
        from . import hip_ops

        hip_ops().multi_tensor_adamw(
            params, grads, exp_avg, exp_avg_sq, master if master is not None else [],
            lr, beta1, beta2, eps, weight_decay, bc1, bc2, grad_scale,
        )
        return
    # CPU reference
    work = master if master is not None else params
    gf = [g.float() * grad_scale for g in grads]
    torch._foreach_mul_(exp_avg, beta1)
    torch._foreach_add_(exp_avg, gf, alpha=1.0 - beta1)
    torch._foreach_mul_(exp_avg_sq, beta2)
    torch._foreach_addcmul_(exp_avg_sq, gf, gf, value=1.0 - beta2)
    for p, m, v in zip(work, exp_avg, exp_avg_sq):
        mhat = m / bc1
        vhat = v / bc2
        p.mul_(1.0 - lr * weight_decay)
        p.add_(mhat / (vhat.sqrt() + eps), alpha=-lr)
    if master is not None:
        for p, w in zip(params, master):
            p.copy_(w.to(p.dtype))
