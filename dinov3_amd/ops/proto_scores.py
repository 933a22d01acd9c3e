"""Prototype-score losses: Sinkhorn-Knopp centering (K17) and the softmax
cross-entropy reductions over the K=65536 prototype axis (K18/K19).

Distributed reductions (C4/C5 in SURVEY §2.3) are RCCL all-reduces on the
K-vector of row sums and the scalar sums — latency-bound small collectives.

Semantics follow the reference exactly:
- dino sinkhorn: dinov3_jax/loss/dino_clstoken_loss.py:35-62
- ibot sinkhorn (B = global n_masked): ibot_patch_loss.py:77-109
- CE: dino_clstoken_loss.py:66-89, ibot lossfunc :13-14
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F


def _world() -> int:
    return dist.get_world_size() if (dist.is_available() and dist.is_initialized()) else 1


@torch.no_grad()
def sinkhorn_knopp(teacher_logits: torch.Tensor, teacher_temp: float,
                   total_columns: Optional[torch.Tensor] = None, n_iterations: int = 3) -> torch.Tensor:
    """Doubly-stochastic teacher targets.

    teacher_logits: [M, K] (M = local rows: batch or masked patches).
    total_columns: scalar tensor = GLOBAL effective column count B. For the
    DINO loss this is local_M * world (computed here when None); for iBOT it
    is the all-reduced n_masked_patches which the CALLER must pre-reduce.
    Returns [M, K] target probabilities (rows sum to ~1 per column scaling).
    """
    world = _world()
    Q = torch.exp(teacher_logits.float() / teacher_temp).T  # [K, M]
    K, M = Q.shape
    if total_columns is None:
        B = torch.tensor(float(M * world), device=Q.device)
    else:
        B = total_columns.float()

    sum_Q = Q.sum()
    if world > 1:
        dist.all_reduce(sum_Q)
    Q /= sum_Q

    for _ in range(n_iterations):
        sum_rows = Q.sum(dim=1, keepdim=True)  # [K, 1]
        if world > 1:
            dist.all_reduce(sum_rows)
        Q /= sum_rows
        Q /= K
        Q /= Q.sum(dim=0, keepdim=True)
        Q /= B

    Q *= B
    return Q.T.contiguous()


def sinkhorn_rowcol(Q: torch.Tensor) -> torch.Tensor:
    """Placeholder hook for the fused row/col-normalization kernel."""
    return Q


def dino_softmax_ce(student_logits: torch.Tensor, teacher_probs: torch.Tensor,
                    student_temp: float = 0.1, ignore_diagonal: bool = False) -> torch.Tensor:
    """student_logits: [S, B, K]; teacher_probs: [T, B, K].

    loss[s,t] = -sum_bk logp_s[s,b,k] * t[t,b,k]; mean over all (s,t,b) pairs,
    optionally zeroing the s==t diagonal (global-vs-global self pairs).
    """
    S, B, K = student_logits.shape
    T = teacher_probs.shape[0]
    logp = F.log_softmax(student_logits.float() / student_temp, dim=-1)
    tp = teacher_probs.float()
    if ignore_diagonal:
        loss_st = -torch.einsum("sbk,tbk->st", logp, tp)
        loss_st = loss_st - torch.diag_embed(torch.diagonal(loss_st))
        M = min(S, T)
        return loss_st.sum() / (B * S * T - B * M)
    return -torch.einsum("sbk,tbk->", logp, tp) / (B * S * T)


def ibot_softmax_ce(student_patch_logits: torch.Tensor, teacher_patch_probs: torch.Tensor,
                    n_total_rows: int, student_temp: float = 0.1,
                    masks_weight: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Masked-patch CE: rows are the gathered masked patches [M, K].

    Meta's DINOv3 weights each row by 1/(patches masked in its sample)
    (masks_weight); the reference drops the weighting — we
    follow Meta when masks_weight is provided, and divide by `n_total_rows`
    (the mask-batch row count) as both implementations do.
    """
    logp = F.log_softmax(student_patch_logits.float() / student_temp, dim=-1)
    per_row = (teacher_patch_probs.float() * logp).sum(dim=-1)  # [M]
    if masks_weight is not None:
        per_row = per_row * masks_weight.float()
        return -per_row.sum()
    return -per_row.sum() / max(n_total_rows, 1)
