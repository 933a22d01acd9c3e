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
    # training-collective scope: the multi-distillation subgroup when set
    from .. import parallel

    return parallel.subgroup_size()


def _group():
    from .. import parallel

    return parallel.subgroup()


class FactoredProbs:
    """Lazy sinkhorn output: t = exp(logits/temp) * u[..., None] * v.

    Sinkhorn-Knopp is diagonal scaling of exp(x/T); carrying only the row/col
    scale vectors lets the CE kernels consume the teacher distribution without
    ever materializing the [rows, 65536] fp32 tensor.
    """

    def __init__(self, logits: torch.Tensor, u: torch.Tensor, v: torch.Tensor, temp: float):
        self.logits = logits
        self.u = u
        self.v = v
        self.temp = temp

    @property
    def shape(self):
        return self.logits.shape

    def reshape(self, *shape):
        return FactoredProbs(self.logits.reshape(*shape), self.u.reshape(*shape[:-1]),
                             self.v, self.temp)

    def __getitem__(self, idx):
        return FactoredProbs(self.logits[idx], self.u[idx], self.v, self.temp)

    def materialize(self) -> torch.Tensor:
        t = torch.exp(self.logits.float() / self.temp)
        return t * self.u.unsqueeze(-1) * self.v

    def float(self):
        return self.materialize()


@torch.no_grad()
def sinkhorn_knopp_factored(teacher_logits: torch.Tensor, teacher_temp: float,
                            n_iterations: int = 3) -> FactoredProbs:
    """Factored sinkhorn: 2 bf16 read passes per iteration, no Q tensor.

    Per-iteration scalar factors (sum_Q, K, B) cancel in the final
    row-normalize, so only the doubly-stochastic scaling vectors survive —
    identical output to the materialized reference."""
    from . import hip_ops

    ops = hip_ops()
    world = _world()
    M, K = teacher_logits.shape
    x = teacher_logits.contiguous()
    u = torch.empty(0)
    for it in range(n_iterations):
        A = ops.sinkhorn_fact_colsum(x, u if it > 0 else torch.empty(0, device=x.device),
                                     teacher_temp)
        if world > 1:
            dist.all_reduce(A, group=_group())
        v = torch.reciprocal(A * K)
        u = ops.sinkhorn_fact_rowsum(x, v, teacher_temp)
    return FactoredProbs(x, u, v, teacher_temp)


class _DinoCEFactFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, xt, u, v, temps, tempt, ignore_diag, denom):
        from . import hip_ops

        loss_sum, lse, st = hip_ops().dino_ce_fact_fwd(x, xt, u, v, temps, tempt, ignore_diag)
        ctx.save_for_backward(x, xt, u, v, lse, st)
        ctx.meta = (temps, tempt, ignore_diag, denom)
        return loss_sum / denom

    @staticmethod
    def backward(ctx, g):
        from . import hip_ops

        x, xt, u, v, lse, st = ctx.saved_tensors
        temps, tempt, ignore_diag, denom = ctx.meta
        gs = (g / denom).float().contiguous()
        dx = hip_ops().dino_ce_fact_bwd(gs, x, xt, u, v, lse, st, temps, tempt, ignore_diag)
        return dx, None, None, None, None, None, None, None


class _IbotCEFactFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, xt, u, v, w, temps, tempt):
        from . import hip_ops

        loss_sum, lse, st = hip_ops().ibot_ce_fact_fwd(x, xt, u, v, w, temps, tempt)
        ctx.save_for_backward(x, xt, u, v, w, lse, st)
        ctx.meta = (temps, tempt)
        return loss_sum

    @staticmethod
    def backward(ctx, g):
        from . import hip_ops

        x, xt, u, v, w, lse, st = ctx.saved_tensors
        temps, tempt = ctx.meta
        dx = hip_ops().ibot_ce_fact_bwd(g.float().contiguous(), x, xt, u, v, w, lse, st,
                                        temps, tempt)
        return dx, None, None, None, None, None, None


class _DinoCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, t, temp, ignore_diag, denom):
        from . import hip_ops

        loss_sum, lse, st = hip_ops().dino_ce_fwd(x, t, temp, ignore_diag)
        ctx.save_for_backward(x, t, lse, st)
        ctx.temp = temp
        ctx.ignore_diag = ignore_diag
        ctx.denom = denom
        return loss_sum / denom

    @staticmethod
    def backward(ctx, g):
        from . import hip_ops

        x, t, lse, st = ctx.saved_tensors
        gs = (g / ctx.denom).float().contiguous()
        dx = hip_ops().dino_ce_bwd(gs, x, t, lse, st, ctx.temp, ctx.ignore_diag)
        return dx, None, None, None, None


class _IbotCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, t, w, temp, denom):
        from . import hip_ops

        loss_sum, lse, st = hip_ops().ibot_ce_fwd(x, t, w, temp)
        ctx.save_for_backward(x, t, w, lse, st)
        ctx.temp = temp
        ctx.denom = denom
        return loss_sum / denom

    @staticmethod
    def backward(ctx, g):
        from . import hip_ops

        x, t, w, lse, st = ctx.saved_tensors
        gs = (g / ctx.denom).float().contiguous()
        dx = hip_ops().ibot_ce_bwd(gs, x, t, w, lse, st, ctx.temp)
        return dx, None, None, None, None


@torch.no_grad()
def _sinkhorn_knopp_hip(teacher_logits: torch.Tensor, teacher_temp: float,
                        total_columns: Optional[torch.Tensor], n_iterations: int) -> torch.Tensor:
    """Fused sinkhorn on [M, K] (reference transposes to [K, M]; we keep the
    row-major layout and swap the reduction roles). The /sum_Q normalization
    folds into the first column-sum divisor — it cancels in the first
    row-normalize, so Q itself is never globally divided."""
    from . import hip_ops

    ops = hip_ops()
    world = _world()
    M, K = teacher_logits.shape
    Q, total = ops.sinkhorn_exp(teacher_logits.contiguous(), teacher_temp)
    if world > 1:
        dist.all_reduce(total, group=_group())
    if total_columns is None:
        B = torch.full((), float(M * world), device=Q.device)
    else:
        B = total_columns.float().reshape(()).to(Q.device)
    one = torch.ones((), device=Q.device)
    for it in range(n_iterations):
        col = ops.sinkhorn_colsum(Q, total if it == 0 else one)
        if world > 1:
            dist.all_reduce(col, group=_group())
        ops.sinkhorn_div_row(Q, col, float(K), B, it == n_iterations - 1)
    return Q


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
    from . import use_hip

    if use_hip(teacher_logits) and teacher_logits.dtype == torch.bfloat16:
        return _sinkhorn_knopp_hip(teacher_logits, teacher_temp, total_columns, n_iterations)
    world = _world()
    Q = torch.exp(teacher_logits.float() / teacher_temp).T  # [K, M]
    K, M = Q.shape
    if total_columns is None:
        B = torch.tensor(float(M * world), device=Q.device)
    else:
        B = total_columns.float()

    sum_Q = Q.sum()
    if world > 1:
        dist.all_reduce(sum_Q, group=_group())
    Q /= sum_Q

    for _ in range(n_iterations):
        sum_rows = Q.sum(dim=1, keepdim=True)  # [K, 1]
        if world > 1:
            dist.all_reduce(sum_rows, group=_group())
        Q /= sum_rows
        Q /= K
        Q /= Q.sum(dim=0, keepdim=True)
        Q /= B

    Q *= B
    return Q.T.contiguous()


def dino_softmax_ce(student_logits: torch.Tensor, teacher_probs: torch.Tensor,
                    student_temp: float = 0.1, ignore_diagonal: bool = False) -> torch.Tensor:
    """student_logits: [S, B, K]; teacher_probs: [T, B, K].

    loss[s,t] = -sum_bk logp_s[s,b,k] * t[t,b,k]; mean over all (s,t,b) pairs,
    optionally zeroing the s==t diagonal (global-vs-global self pairs).
    """
    from . import use_hip

    S, B, K = student_logits.shape
    T = teacher_probs.shape[0]
    if use_hip(student_logits) and student_logits.dtype == torch.bfloat16:
        denom = float(B * S * T - B * min(S, T)) if ignore_diagonal else float(B * S * T)
        if isinstance(teacher_probs, FactoredProbs):
            return _DinoCEFactFn.apply(
                student_logits.contiguous(), teacher_probs.logits.contiguous(),
                teacher_probs.u.reshape(-1).contiguous(), teacher_probs.v,
                student_temp, teacher_probs.temp, ignore_diagonal, denom,
            )
        return _DinoCEFn.apply(
            student_logits.contiguous(), teacher_probs.float().contiguous(),
            student_temp, ignore_diagonal, denom,
        )
    if isinstance(teacher_probs, FactoredProbs):
        teacher_probs = teacher_probs.materialize()
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
    from . import use_hip

    if (use_hip(student_patch_logits) and student_patch_logits.dtype == torch.bfloat16
            and student_patch_logits.shape[0] > 0):
        M = student_patch_logits.shape[0]
        w = (masks_weight.float().contiguous() if masks_weight is not None
             else torch.full((M,), 1.0 / max(n_total_rows, 1),
                             device=student_patch_logits.device))
        if isinstance(teacher_patch_probs, FactoredProbs):
            return _IbotCEFactFn.apply(
                student_patch_logits.contiguous(),
                teacher_patch_probs.logits.contiguous(),
                teacher_patch_probs.u.contiguous(), teacher_patch_probs.v, w,
                student_temp, teacher_patch_probs.temp,
            )
        denom = 1.0  # weights carry the normalization; caller divides by rows
        return _IbotCEFn.apply(
            student_patch_logits.contiguous(), teacher_patch_probs.float().contiguous(),
            w, student_temp, denom,
        )
    if isinstance(teacher_patch_probs, FactoredProbs):
        teacher_patch_probs = teacher_patch_probs.materialize()
    logp = F.log_softmax(student_patch_logits.float() / student_temp, dim=-1)
    per_row = (teacher_patch_probs.float() * logp).sum(dim=-1)  # [M]
    if masks_weight is not None:
        per_row = per_row * masks_weight.float()
        return -per_row.sum()
    return -per_row.sum() / max(n_total_rows, 1)
