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


_KOLEO_GROUP_CACHE = {}


def _koleo_group(ranks_per_group: int, adjacent: bool):
    """Process subgroup for the KoLeo gather. Lazily created ONCE per layout
    (dist.new_group is collective — every rank reaches this in the same order
    through the identical loss code path). Returns (group, group_rank)."""
    world = dist.get_world_size()
    rank = dist.get_rank()
    key = (ranks_per_group, adjacent)
    if key not in _KOLEO_GROUP_CACHE:
        my_group, my_group_rank = None, 0
        n_groups = world // ranks_per_group
        for g in range(n_groups):
            if adjacent:
                ranks = list(range(g * ranks_per_group, (g + 1) * ranks_per_group))
            else:
                ranks = list(range(g, world, n_groups))
            pg = dist.new_group(ranks=ranks)
            if rank in ranks:
                my_group, my_group_rank = pg, ranks.index(rank)
        _KOLEO_GROUP_CACHE[key] = (my_group, my_group_rank)
    return _KOLEO_GROUP_CACHE[key]


class KoLeoLossDistributed(nn.Module):
    """Top-k KoLeo over a gathered batch. `loss_group_size` bounds the
    nearest-neighbor set: it is a SAMPLE count (reference schema
    ssl_default_config.yaml:32 — "If None, uses global batch size"), realized
    as a process subgroup of loss_group_size // local_B ranks. `group_data`
    selects adjacent ranks (same data-distribution slice, the default) vs
    strided ranks (ssl_default_config.yaml:33)."""

    def __init__(self, topk: int = 1, loss_group_size: Optional[int] = None,
                 group_data: bool = True):
        super().__init__()
        self.topk = topk
        self.loss_group_size = loss_group_size
        self.group_data = group_data

    def _gather_group(self, local_B: int):
        """(process group or None for world, group rank, group world size)."""
        world = dist.get_world_size()
        if self.loss_group_size is None:
            return None, dist.get_rank(), world
        ranks_per_group = max(1, min(world, self.loss_group_size // max(local_B, 1)))
        while world % ranks_per_group != 0:  # shrink to a divisor of world
            ranks_per_group -= 1
        if ranks_per_group == world:
            return None, dist.get_rank(), world
        group, group_rank = _koleo_group(ranks_per_group, self.group_data)
        return group, group_rank, ranks_per_group

    def forward(self, student_output: torch.Tensor, eps: float = 1e-8) -> torch.Tensor:
        from .. import parallel

        x = student_output.float()
        x = x / (x.norm(p=2, dim=-1, keepdim=True) + eps)
        local_B = x.shape[0]
        if dist.is_available() and dist.is_initialized() and parallel.subgroup_size() > 1:
            if parallel.subgroup() is not None:
                # multi-distillation: gather over the student's subgroup
                # (loss_group_size sub-partitioning is not composed with it)
                group, rank, gw = (parallel.subgroup(), parallel.subgroup_rank(),
                                   parallel.subgroup_size())
            else:
                group, rank, gw = self._gather_group(local_B)
            gathered = [torch.empty_like(x) for _ in range(gw)]
            dist.all_gather(gathered, x, group=group)
            gathered[rank] = x  # keep autograd path through the local shard
            all_x = torch.cat(gathered, dim=0)
        else:
            rank = 0
            all_x = x
        dots = x @ all_x.T
        rows = torch.arange(local_B, device=x.device)
        dots[rows, rank * local_B + rows] = -1.0
        indices = dots.topk(self.topk, dim=1).indices  # [local_B, topk]
        x_expanded = x.repeat_interleave(self.topk, dim=0)
        neighbours = all_x[indices.flatten()]
        distances = pairwise_distance(x_expanded, neighbours, eps)
        return -torch.log(distances + eps).mean()
