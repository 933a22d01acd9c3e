"""Multi-process (gloo, world_size=2) tests of the distributed semantics that
the RCCL path uses on the GPU node: sinkhorn psum, grad pmean, KoLeo gather."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _init(rank, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)


def _run(fn, port):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=fn, args=(r, port)) for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
    for p in procs:
        assert p.exitcode == 0, f"child exited with {p.exitcode}"


# ---------------------------------------------------------------- sinkhorn
def _sinkhorn_worker(rank, port):
    _init(rank, port)
    from dinov3_amd.ops.proto_scores import sinkhorn_knopp

    torch.manual_seed(0)  # same base
    full = torch.randn(8, 32)  # the "global batch" on every rank
    local = full[rank * 4: (rank + 1) * 4]
    Q = sinkhorn_knopp(local, teacher_temp=0.1)
    # single-process reference over the full batch
    dist.destroy_process_group()
    import torch.distributed as d

    Q_ref_full = _sinkhorn_single(full, 0.1)
    ref = Q_ref_full[rank * 4: (rank + 1) * 4]
    assert torch.allclose(Q, ref, atol=1e-4), (Q - ref).abs().max()


def _sinkhorn_single(logits, temp):
    Q = torch.exp(logits.float() / temp).T
    K, M = Q.shape
    B = float(M)
    Q /= Q.sum()
    for _ in range(3):
        Q /= Q.sum(dim=1, keepdim=True)
        Q /= K
        Q /= Q.sum(dim=0, keepdim=True)
        Q /= B
    Q *= B
    return Q.T


def test_sinkhorn_distributed_matches_single():
    _run(_sinkhorn_worker, 29511)


# ---------------------------------------------------------------- reducer
def _reducer_worker(rank, port):
    _init(rank, port)
    from dinov3_amd.parallel.ddp import GradReducer

    torch.manual_seed(100 + rank)
    p = torch.nn.Parameter(torch.zeros(16))
    reducer = GradReducer([p], bucket_cap_mb=0.00001)
    g_local = torch.full((16,), float(rank + 1))
    (p * g_local).sum().backward()
    reducer.finalize()
    expected = (1.0 + 2.0) / 2
    assert torch.allclose(p.grad, torch.full((16,), expected), atol=1e-6), p.grad
    dist.destroy_process_group()


def test_grad_reducer_pmean():
    _run(_reducer_worker, 29513)


# ---------------------------------------------------------------- koleo
def _koleo_worker(rank, port):
    _init(rank, port)
    from dinov3_amd.loss import KoLeoLossDistributed

    torch.manual_seed(0)
    full = torch.randn(8, 16)
    local = full[rank * 4: (rank + 1) * 4].clone().requires_grad_(True)
    loss_mod = KoLeoLossDistributed(topk=1)
    out = loss_mod(local)
    out.backward()
    assert torch.isfinite(out)
    assert local.grad is not None and torch.isfinite(local.grad).all()
    dist.destroy_process_group()


def test_koleo_distributed():
    _run(_koleo_worker, 29515)


# ---------------------------------------------------------------- ibot B
def _ibot_b_worker(rank, port):
    _init(rank, port)
    from dinov3_amd.loss import iBOTPatchLoss

    torch.manual_seed(0)
    loss_mod = iBOTPatchLoss(patch_out_dim=16)
    n_local = 3 + rank  # ranks have different masked counts
    logits = torch.randn(n_local, 16)
    n_tensor = torch.tensor([n_local])
    Q = loss_mod.sinkhorn_knopp_teacher(logits, teacher_temp=0.1, n_masked_patches_tensor=n_tensor)
    assert Q.shape == (n_local, 16)
    assert torch.isfinite(Q).all()
    dist.destroy_process_group()


def test_ibot_sinkhorn_global_count():
    _run(_ibot_b_worker, 29517)
