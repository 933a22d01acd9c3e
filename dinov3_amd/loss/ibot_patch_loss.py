"""iBOT masked-patch loss (parity: dinov3_jax/loss/ibot_patch_loss.py:18-109).

The reference drops the per-sample masks_weight; we follow
Meta's DINOv3 semantics and apply it.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from ..ops.proto_scores import ibot_softmax_ce, sinkhorn_knopp


class iBOTPatchLoss(nn.Module):
    def __init__(self, patch_out_dim: int, student_temp: float = 0.1, center_momentum: float = 0.9):
        super().__init__()
        self.student_temp = student_temp
        self.center_momentum = center_momentum
        self.register_buffer("center", torch.zeros(1, 1, patch_out_dim))

    @torch.no_grad()
    def softmax_center_teacher(self, teacher_patch_tokens: torch.Tensor, teacher_temp: float,
                               update_centers: bool = True) -> torch.Tensor:
        if update_centers:
            self.apply_center_update(teacher_patch_tokens)
        return F.softmax((teacher_patch_tokens.float() - self.center) / teacher_temp, dim=-1)

    @torch.no_grad()
    def sinkhorn_knopp_teacher(self, teacher_output: torch.Tensor, teacher_temp: float,
                               n_masked_patches_tensor: torch.Tensor, n_iterations: int = 3):
        from ..ops import use_hip
        from ..ops.proto_scores import sinkhorn_knopp_factored

        if use_hip(teacher_output) and teacher_output.dtype == torch.bfloat16:
            # factored form (the per-iteration B scalar cancels in the final
            # row-normalize, so the global masked count is not needed)
            return sinkhorn_knopp_factored(teacher_output, teacher_temp, n_iterations)
        B = n_masked_patches_tensor.clone().float()
        if B.ndim > 0:
            B = B.sum()
        if dist.is_available() and dist.is_initialized():
            from .. import parallel

            B = B.to(teacher_output.device)
            dist.all_reduce(B, group=parallel.subgroup())
        return sinkhorn_knopp(teacher_output, teacher_temp, total_columns=B, n_iterations=n_iterations)

    def forward(self, student_patch_tokens: torch.Tensor, teacher_patch_tokens: torch.Tensor,
                student_masks_flat: torch.Tensor) -> torch.Tensor:
        """Unmasked-layout variant: [B, N, K] inputs, boolean mask [B, N]."""
        logp = F.log_softmax(student_patch_tokens.float() / self.student_temp, dim=-1)
        loss = (teacher_patch_tokens.float() * logp).sum(dim=-1)  # [B, N]
        m = student_masks_flat.float()
        per_sample = (loss * m).sum(dim=-1) / m.sum(dim=-1).clamp(1.0)
        return -per_sample.mean()

    def forward_masked(
        self,
        student_patch_tokens_masked: torch.Tensor,
        teacher_patch_tokens_masked: torch.Tensor,
        student_masks_flat: torch.Tensor,
        n_masked_patches: Optional[int] = None,
        masks_weight: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        s = student_patch_tokens_masked
        t = teacher_patch_tokens_masked
        if masks_weight is None:
            weights = 1.0 / student_masks_flat.sum(dim=-1).clamp(1.0)
            masks_weight = torch.where(student_masks_flat, weights.unsqueeze(-1),
                                       torch.zeros_like(weights).unsqueeze(-1))
            masks_weight = masks_weight[student_masks_flat]
        if n_masked_patches is not None:
            s = s[:n_masked_patches]
            t = t[:n_masked_patches]
            masks_weight = masks_weight[:n_masked_patches]
        # -sum(weighted per-row CE) / (rows in the mask batch)
        return ibot_softmax_ce(
            s, t, n_total_rows=student_masks_flat.shape[0],
            student_temp=self.student_temp, masks_weight=masks_weight,
        ) / student_masks_flat.shape[0]

    @torch.no_grad()
    def apply_center_update(self, teacher_output: torch.Tensor) -> None:
        from .. import parallel

        local_center = teacher_output.float().mean(dim=0, keepdim=True)
        if dist.is_available() and dist.is_initialized():
            dist.all_reduce(local_center, group=parallel.subgroup())
            local_center /= parallel.subgroup_size()
        self.center.mul_(self.center_momentum).add_(local_center * (1 - self.center_momentum))
