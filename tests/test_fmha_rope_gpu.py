"""GPU numerics for the fused rope+FMHA-on-qkv path (ops/flat_attention.py)."""

import math
import os

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _ref_group(qkv, sin, cos, prefix):
    """fp32 torch reference on [B,N,3,H,hd]."""
    B, N, _, H, hd = qkv.shape
    q, k, v = qkv.float().permute(2, 0, 3, 1, 4).unbind(0)  # [B,H,N,hd]
    if sin is not None:
        half = hd // 2

        def rot(x):
            head = x[..., :prefix, :]
            tail = x[..., prefix:, :]
            r = torch.cat([-tail[..., half:], tail[..., :half]], dim=-1)
            return torch.cat([head, tail * cos + r * sin], dim=-2)

        q, k = rot(q), rot(k)
    s = torch.einsum("bhqd,bhkd->bhqk", q, k) / math.sqrt(hd)
    p = torch.softmax(s, dim=-1)
    o = torch.einsum("bhqk,bhkd->bhqd", p, v)
    return o.permute(0, 2, 1, 3)  # [B,N,H,hd]


@pytest.mark.parametrize("N,hd,prefix", [(37, 64, 1), (197, 64, 1), (197, 64, 5), (64, 64, 0), (197, 128, 1), (300, 64, 2)])
def test_fmha_rope_fwd(N, hd, prefix):
    from dinov3_amd.ops import hip_ops

    torch.manual_seed(0)
    B, H = 2, 4
    qkv = torch.randn(B, N, 3, H, hd, device=DEV).bfloat16()
    P = N - prefix
    angles = torch.rand(P, hd // 2, device=DEV) * 6.28
    angles = torch.cat([angles, angles], dim=-1)
    sin, cos = angles.sin().contiguous(), angles.cos().contiguous()
    o, lse = hip_ops().fmha_rope_fwd(qkv, sin, cos, prefix)
    ref = _ref_group(qkv, sin, cos, prefix)
    err = (o.float() - ref).abs().max().item()
    assert err < 0.03, f"fwd err {err}"
    assert torch.isfinite(lse).all()


def test_fmha_rope_no_rope():
    from dinov3_amd.ops import hip_ops

    torch.manual_seed(1)
    B, H, N, hd = 2, 3, 101, 64
    qkv = torch.randn(B, N, 3, H, hd, device=DEV).bfloat16()
    e = torch.empty(0, device=DEV)
    o, lse = hip_ops().fmha_rope_fwd(qkv, e, e, 0)
    ref = _ref_group(qkv, None, None, 0)
    err = (o.float() - ref).abs().max().item()
    assert err < 0.03, f"fwd err {err}"


@pytest.mark.parametrize("N,hd,prefix", [(37, 64, 1), (197, 64, 1), (197, 128, 1), (54, 128, 5), (64, 128, 0)])
def test_fmha_rope_bwd(N, hd, prefix):
    from dinov3_amd.ops.flat_attention import flat_multi_fmha

    torch.manual_seed(2)
    B, H = 2, 3
    D = H * hd
    qkv_flat = torch.randn(B * N, 3 * D, device=DEV).bfloat16().requires_grad_(True)
    P = N - prefix
    angles = torch.rand(P, hd // 2, device=DEV) * 6.28
    angles = torch.cat([angles, angles], dim=-1)
    sin, cos = angles.sin().contiguous(), angles.cos().contiguous()
    metas = [(0, B, N, sin, cos, prefix)]
    out = flat_multi_fmha(qkv_flat, H, metas)
    dy = torch.randn_like(out)
    out.backward(dy)

    ref_in = qkv_flat.detach().float().requires_grad_(True)
    qkv_g = ref_in.view(B, N, 3, H, hd)
    o_ref = _ref_group(qkv_g, sin, cos, prefix).reshape(B * N, D)
    o_ref.backward(dy.float().reshape(B * N, D))
    err_o = (out.float() - o_ref).abs().max().item()
    err_g = (qkv_flat.grad.float() - ref_in.grad).abs().max().item()
    scale = ref_in.grad.abs().max().item()
    assert err_o < 0.03, f"fwd err {err_o}"
    assert err_g < 0.02 + 0.02 * scale, f"bwd err {err_g} (scale {scale})"


def test_flat_multi_group():
    """Two crop groups through one flat buffer == per-group reference."""
    from dinov3_amd.ops.flat_attention import flat_multi_fmha

    torch.manual_seed(3)
    H, hd = 4, 64
    D = H * hd
    B1, N1, B2, N2 = 2, 197, 4, 37
    qkv_flat = torch.randn(B1 * N1 + B2 * N2, 3 * D, device=DEV).bfloat16()
    metas = []
    tabs = []
    for off, B, N in ((0, B1, N1), (B1 * N1, B2, N2)):
        P = N - 1
        a = torch.rand(P, hd // 2, device=DEV) * 6.28
        a = torch.cat([a, a], dim=-1)
        sin, cos = a.sin().contiguous(), a.cos().contiguous()
        metas.append((off, B, N, sin, cos, 1))
        tabs.append((sin, cos))
    out = flat_multi_fmha(qkv_flat, H, metas)
    for (off, B, N, sin, cos, prefix) in metas:
        ref = _ref_group(qkv_flat[off: off + B * N].view(B, N, 3, H, hd), sin, cos, prefix)
        got = out[off: off + B * N].view(B, N, H, hd).float()
        err = (got - ref).abs().max().item()
        assert err < 0.03, f"group err {err}"


@pytest.mark.parametrize("N,hd,prefix", [(197, 64, 1), (197, 128, 1), (300, 64, 2), (128, 64, 0)])
def test_dkv64_variant(N, hd, prefix, monkeypatch):
    """Gated 64-row super-tile dkv kernel vs the fp32 reference (the round-1
    attempt NaN'd from a 32-row LDS stride; this variant fixes the staging
    layout — flip DINOV3_FMHA_DKV64 on by default once this passes)."""
    monkeypatch.setenv("DINOV3_FMHA_DKV64", "1")
    from dinov3_amd.ops.flat_attention import flat_multi_fmha

    torch.manual_seed(4)
    B, H = 2, 3
    D = H * hd
    qkv_flat = torch.randn(B * N, 3 * D, device=DEV).bfloat16().requires_grad_(True)
    P = N - prefix
    angles = torch.rand(P, hd // 2, device=DEV) * 6.28
    angles = torch.cat([angles, angles], dim=-1)
    sin, cos = angles.sin().contiguous(), angles.cos().contiguous()
    out = flat_multi_fmha(qkv_flat, H, [(0, B, N, sin, cos, prefix)])
    dy = torch.randn_like(out)
    out.backward(dy)

    ref_in = qkv_flat.detach().float().requires_grad_(True)
    o_ref = _ref_group(ref_in.view(B, N, 3, H, hd), sin, cos, prefix).reshape(B * N, D)
    o_ref.backward(dy.float().reshape(B * N, D))
    err = (qkv_flat.grad.float() - ref_in.grad).abs().max().item()
    scale = ref_in.grad.abs().max().item()
    assert torch.isfinite(qkv_flat.grad).all(), "dkv64: non-finite grads"
    assert err < 0.02 + 0.02 * scale, f"dkv64 bwd err {err} (scale {scale})"
