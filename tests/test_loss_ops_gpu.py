"""GPU numerics for the fused prototype-CE and sinkhorn kernels."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _close(a, b, atol, what=""):
    err = (a.float() - b.float()).abs().max().item()
    assert err <= atol, f"{what}: max err {err}"


@pytest.mark.parametrize("ignore_diag", [False, True])
def test_dino_ce_fused_matches_reference(ignore_diag):
    torch.manual_seed(0)
    S, T, B, K = 4, 2, 6, 4096
    if ignore_diag:
        S = T = 2
    x = torch.randn(S, B, K, device=DEV).bfloat16().requires_grad_(True)
    t = torch.softmax(torch.randn(T, B, K, device=DEV), dim=-1)
    from dinov3_amd.ops.proto_scores import dino_softmax_ce

    loss = dino_softmax_ce(x, t, student_temp=0.1, ignore_diagonal=ignore_diag)
    loss.backward()

    xr = x.detach().float().requires_grad_(True)
    logp = F.log_softmax(xr / 0.1, dim=-1)
    if ignore_diag:
        lst = -torch.einsum("sbk,tbk->st", logp, t)
        lst = lst - torch.diag_embed(torch.diagonal(lst))
        ref = lst.sum() / (B * S * T - B * min(S, T))
    else:
        ref = -torch.einsum("sbk,tbk->", logp, t) / (B * S * T)
    ref.backward()
    _close(loss, ref, 5e-3, "dino ce loss")
    # grads are returned in the input dtype (bf16): compare vs quantized ref
    _close(x.grad, xr.grad.bfloat16(), 1e-4 + 0.01 * xr.grad.abs().max().item(), "dino ce dx")


def test_ibot_ce_fused_matches_reference():
    torch.manual_seed(1)
    M, K = 57, 8192
    x = torch.randn(M, K, device=DEV).bfloat16().requires_grad_(True)
    t = torch.softmax(torch.randn(M, K, device=DEV), dim=-1)
    w = torch.rand(M, device=DEV) + 0.1
    from dinov3_amd.ops.proto_scores import ibot_softmax_ce

    loss = ibot_softmax_ce(x, t, n_total_rows=M, student_temp=0.1, masks_weight=w)
    loss.backward()

    xr = x.detach().float().requires_grad_(True)
    logp = F.log_softmax(xr / 0.1, dim=-1)
    ref = -((t * logp).sum(-1) * w).sum()
    ref.backward()
    _close(loss, ref, max(5e-3, 1e-4 * ref.abs().item()), "ibot ce loss")
    _close(x.grad, xr.grad.bfloat16(), 1e-4 + 0.01 * xr.grad.abs().max().item(), "ibot ce dx")


def test_sinkhorn_fused_matches_torch():
    torch.manual_seed(2)
    M, K = 48, 2048
    logits = (torch.randn(M, K, device=DEV) * 2).bfloat16()
    import os

    from dinov3_amd.ops.proto_scores import sinkhorn_knopp

    Q_hip = sinkhorn_knopp(logits, teacher_temp=0.1)
    os.environ["DINOV3_DISABLE_HIP"] = "1"
    try:
        Q_ref = sinkhorn_knopp(logits.float(), teacher_temp=0.1)
    finally:
        del os.environ["DINOV3_DISABLE_HIP"]
    _close(Q_hip, Q_ref, 2e-3, "sinkhorn")
    _close(Q_hip.sum(dim=1), torch.ones(M, device=DEV), 1e-3, "sinkhorn rowsum")


def test_sinkhorn_fused_ibot_total_columns():
    torch.manual_seed(3)
    M, K = 33, 1024
    logits = torch.randn(M, K, device=DEV).bfloat16()
    total = torch.tensor([float(M)], device=DEV)
    import os

    from dinov3_amd.ops.proto_scores import sinkhorn_knopp

    Q_hip = sinkhorn_knopp(logits, teacher_temp=0.07, total_columns=total)
    os.environ["DINOV3_DISABLE_HIP"] = "1"
    try:
        Q_ref = sinkhorn_knopp(logits.float(), teacher_temp=0.07, total_columns=total)
    finally:
        del os.environ["DINOV3_DISABLE_HIP"]
    _close(Q_hip, Q_ref, 2e-3, "sinkhorn ibot")


def test_sinkhorn_factored_matches_materialized():
    torch.manual_seed(5)
    M, K = 40, 4096
    logits = torch.randn(M, K, device=DEV).bfloat16()
    from dinov3_amd.ops.proto_scores import sinkhorn_knopp, sinkhorn_knopp_factored

    fact = sinkhorn_knopp_factored(logits, teacher_temp=0.1)
    Q_fact = fact.materialize()
    import os

    os.environ["DINOV3_DISABLE_HIP"] = "1"
    try:
        Q_ref = sinkhorn_knopp(logits.float(), teacher_temp=0.1)
    finally:
        del os.environ["DINOV3_DISABLE_HIP"]
    _close(Q_fact, Q_ref, 2e-3, "sinkhorn factored vs materialized")
    _close(Q_fact.sum(dim=1), torch.ones(M, device=DEV), 1e-3, "factored rowsum")


def test_dino_ce_factored_matches_dense():
    torch.manual_seed(6)
    S, T, B, K = 3, 2, 4, 2048
    x = torch.randn(S, B, K, device=DEV).bfloat16().requires_grad_(True)
    xt = torch.randn(T, B, K, device=DEV).bfloat16()
    from dinov3_amd.ops.proto_scores import (FactoredProbs, dino_softmax_ce,
                                             sinkhorn_knopp_factored)

    fact = sinkhorn_knopp_factored(xt.reshape(T * B, K), 0.07)
    fact3 = fact.reshape(T, B, K)
    loss = dino_softmax_ce(x, fact3, student_temp=0.1)
    loss.backward()
    g_fact = x.grad.clone()
    x.grad = None
    dense = fact3.materialize()
    loss_dense = dino_softmax_ce(x, dense, student_temp=0.1)
    loss_dense.backward()
    _close(loss, loss_dense, 5e-3, "dino fact loss")
    _close(g_fact, x.grad, 1e-4 + 0.02 * x.grad.abs().max().item(), "dino fact grad")


def test_ibot_ce_factored_matches_dense():
    torch.manual_seed(7)
    M, K = 30, 2048
    x = torch.randn(M, K, device=DEV).bfloat16().requires_grad_(True)
    xt = torch.randn(M, K, device=DEV).bfloat16()
    w = torch.rand(M, device=DEV) + 0.1
    from dinov3_amd.ops.proto_scores import ibot_softmax_ce, sinkhorn_knopp_factored

    fact = sinkhorn_knopp_factored(xt, 0.07)
    loss = ibot_softmax_ce(x, fact, n_total_rows=M, student_temp=0.1, masks_weight=w)
    loss.backward()
    g_fact = x.grad.clone()
    x.grad = None
    dense = fact.materialize()
    loss_dense = ibot_softmax_ce(x, dense, n_total_rows=M, student_temp=0.1, masks_weight=w)
    loss_dense.backward()
    _close(loss, loss_dense, max(5e-3, 1e-4 * abs(loss_dense.item())), "ibot fact loss")
    _close(g_fact, x.grad, 1e-4 + 0.02 * x.grad.abs().max().item(), "ibot fact grad")
