"""DINO CLS-token loss with Sinkhorn-Knopp or EMA centering.

Parity: dinov3_jax/loss/dino_clstoken_loss.py:14-95. Cross-device reductions
(C4/C7 in SURVEY §2.3) are RCCL all-reduces.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from ..ops.proto_scores import dino_softmax_ce, sinkhorn_knopp


class DINOLoss(nn.Module):
    def __init__(self, out_dim: int, student_temp: float = 0.1, center_momentum: float = 0.9):
        super().__init__()
        self.student_temp = student_temp
        self.center_momentum = center_momentum
        self.register_buffer("center", torch.zeros(1, out_dim))

    @torch.no_grad()
    def softmax_center_teacher(self, teacher_output: torch.Tensor, teacher_temp: float,
                               update_centers: bool = True) -> torch.Tensor:
        if update_centers:
            self.apply_center_update(teacher_output)
        return F.softmax((teacher_output.float() - self.center) / teacher_temp, dim=-1)

    @torch.no_grad()
    def sinkhorn_knopp_teacher(self, teacher_output: torch.Tensor, teacher_temp: float,
                               n_iterations: int = 3):
        from ..ops import use_hip
        from ..ops.proto_scores import sinkhorn_knopp_factored

        if use_hip(teacher_output) and teacher_output.dtype == torch.bfloat16:
            # factored form: the CE kernel consumes exp(x/T)*u*v lazily
            return sinkhorn_knopp_factored(teacher_output, teacher_temp, n_iterations)
        return sinkhorn_knopp(teacher_output, teacher_temp, n_iterations=n_iterations)

    def forward(self, student_logits: torch.Tensor, teacher_probs: torch.Tensor,
                ignore_diagonal: bool = False) -> torch.Tensor:
        """student_logits: [S, B, K]; teacher_probs: [T, B, K]."""
        return dino_softmax_ce(student_logits, teacher_probs, self.student_temp, ignore_diagonal)

    @torch.no_grad()
    def apply_center_update(self, teacher_output: torch.Tensor) -> None:
        from .. import parallel

        local_center = teacher_output.float().mean(dim=0, keepdim=True)
        if dist.is_available() and dist.is_initialized():
            dist.all_reduce(local_center, group=parallel.subgroup())
            local_center /= parallel.subgroup_size()
        self.center.mul_(self.center_momentum).add_(local_center * (1 - self.center_momentum))
