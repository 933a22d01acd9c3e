"""KoLeo regularizer, local and distributed (all-gather over RCCL).

Parity: dinov3_jax/loss/koleo_loss.py:20-69. The distributed variant gathers
cls features across ranks (C6 in SURVEY §2.3) and takes top-k NNs over the
global batch, masking each row's own global position.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn



def pairwise_distance(x: torch.Tensor, y: torch.Tensor, eps: float = 1e-8) -> torch.Tensor:
    return (x - y).norm(p=2, dim=-1) + eps


class KoLeoLoss(nn.Module):
    def pairwise_NNs_inner(self, x: torch.Tensor) -> torch.Tensor:
        dots = x @ x.T
        dots.fill_diagonal_(-1.0)
        return dots.argmax(dim=1)

    def forward(self, student_output: torch.Tensor, eps: float = 1e-8) -> torch.Tensor:
        x = student_output.float()
        x = x / (x.norm(p=2, dim=-1, keepdim=True) + eps)
        indices = self.pairwise_NNs_inner(x)
        distances = pairwise_distance(x, x[indices], eps)
        return -torch.log(distances + eps).mean()


class KoLeoLossDistributed(nn.Module):
    def __init__(self, topk: int = 1, loss_group_size: Optional[int] = None):
        super().__init__()
        self.topk = topk
        self.loss_group_size = loss_group_size

    def forward(self, student_output: torch.Tensor, eps: float = 1e-8) -> torch.Tensor:
        x = student_output.float()
        x = x / (x.norm(p=2, dim=-1, keepdim=True) + eps)
        if dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1:
            world = dist.get_world_size()
            rank = dist.get_rank()
            gathered = [torch.empty_like(x) for _ in range(world)]
            dist.all_gather(gathered, x)
            gathered[rank] = x  # keep autograd path through the local shard
            all_x = torch.cat(gathered, dim=0)
        else:
            rank = 0
            all_x = x
        local_B = x.shape[0]
        dots = x @ all_x.T
        rows = torch.arange(local_B, device=x.device)
        dots[rows, rank * local_B + rows] = -1.0
        indices = dots.topk(self.topk, dim=1).indices  # [local_B, topk]
        x_expanded = x.repeat_interleave(self.topk, dim=0)
        neighbours = all_x[indices.flatten()]
        distances = pairwise_distance(x_expanded, neighbours, eps)
        return -torch.log(distances + eps).mean()
