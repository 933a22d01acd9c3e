"""Gram anchoring loss (parity: dinov3_jax/loss/gram_loss.py:13-51, with the
in-place boolean assignment bug §8 B5 fixed via torch.where)."""

from __future__ import annotations

import torch
import torch.nn as nn


class GramLoss(nn.Module):
    def __init__(self, apply_norm: bool = True, img_level: bool = True,
                 remove_neg: bool = True, remove_only_teacher_neg: bool = False):
        super().__init__()
        assert remove_neg != remove_only_teacher_neg
        self.apply_norm = apply_norm
        self.img_level = img_level
        self.remove_neg = remove_neg
        self.remove_only_teacher_neg = remove_only_teacher_neg

    def forward(self, output_feats: torch.Tensor, target_feats: torch.Tensor,
                img_level: bool = True) -> torch.Tensor:
        out = output_feats.float()
        tgt = target_feats.float()
        if img_level:
            assert out.ndim == 3 and tgt.ndim == 3
        if self.apply_norm:
            tgt = tgt / tgt.norm(dim=-1, keepdim=True)
            out = out / out.norm(dim=-1, keepdim=True)
        if not img_level:
            tgt = tgt.reshape(-1, tgt.shape[-1])
            out = out.reshape(-1, out.shape[-1])
        target_sim = tgt @ tgt.transpose(-1, -2)
        student_sim = out @ out.transpose(-1, -2)
        if self.remove_neg:
            target_sim = torch.where(target_sim < 0, torch.zeros_like(target_sim), target_sim)
            student_sim = torch.where(student_sim < 0, torch.zeros_like(student_sim), student_sim)
        elif self.remove_only_teacher_neg:
            both_neg = (student_sim < 0) & (target_sim < 0)
            student_sim = torch.where(both_neg, torch.zeros_like(student_sim), student_sim)
            target_sim = torch.where(target_sim < 0, torch.zeros_like(target_sim), target_sim)
        return ((student_sim - target_sim) ** 2).mean()
