"""hipBLASLt fused-epilogue numerics vs plain fp32 torch (round-2 path).

Gated on DINOV3_BLASLT_MLP=1 in addition to the gpu marker: the default
driver run skips these until the epilogue path is validated and enabled.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

# hipBLASLt on ROCm 7.2 / gfx950 exposes only BIAS and GELU_BIAS epilogues
# (AUX/DGELU/BGRAD probed on hardware: 0 algorithms) — the fused-MLP tests
# stay gated until a stack ships those epilogues.
needs_aux_epilogues = pytest.mark.skipif(
    os.environ.get("DINOV3_BLASLT_MLP", "0") != "1",
    reason="hipBLASLt aux/bgrad epilogues unsupported on this stack "
           "(set DINOV3_BLASLT_MLP=1 to re-probe)")

DEV = "cuda"


@pytest.fixture(scope="module")
def ops():
    from dinov3_amd.ops import hip_ops

    return hip_ops()


def _close(got, want, atol, what):
    err = (got.float() - want.float()).abs().max().item()
    assert err < atol, f"{what}: max err {err}"


def test_gemm_bias_layout(ops):
    torch.manual_seed(0)
    M, K, N = 192, 256, 320
    x = torch.randn(M, K, device=DEV).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.05).bfloat16()
    b = torch.randn(N, device=DEV).bfloat16()
    y = ops.blaslt_gemm_bias(x, w, b)
    ref = x.float() @ w.float().T + b.float()
    _close(y, ref, 0.25, "blaslt gemm+bias")


def test_epilogue_support_matrix(ops):
    """Pin the probed support matrix: BIAS/GELU_BIAS exist, aux/grad don't.
    If this ever flips on a new ROCm, the fused-MLP path becomes viable."""
    F32, BF16 = 0, 14
    assert ops.blaslt_probe_epilogue(4, -1, BF16, 512, 384, 128) > 0    # BIAS
    assert ops.blaslt_probe_epilogue(36, -1, BF16, 512, 384, 128) > 0   # GELU_BIAS
    aux_like = [(160, BF16, -1), (164, BF16, BF16), (192, BF16, -1),
                (208, BF16, F32), (512, -1, F32)]
    supported = [e for (e, at, bt) in aux_like
                 if ops.blaslt_probe_epilogue(e, at, bt, 512, 384, 128) > 0]
    assert not supported, (
        f"hipBLASLt now supports epilogues {supported} — revisit the fused "
        "MLP path in ops/blaslt_mlp.py")


@needs_aux_epilogues
def test_gemm_bias_gelu_aux(ops):
    torch.manual_seed(1)
    M, K, N = 384, 128, 512
    x = torch.randn(M, K, device=DEV).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.05).bfloat16()
    b = torch.randn(N, device=DEV).bfloat16()
    h, pre = ops.blaslt_gemm_bias_gelu_fwd(x, w, b)
    pre_ref = x.float() @ w.float().T + b.float()
    _close(pre, pre_ref, 0.25, "aux pre-activation")
    _close(h, torch.nn.functional.gelu(pre_ref, approximate="tanh"), 0.25, "gelu output")


@needs_aux_epilogues
def test_gemm_dgelu_bgrad(ops):
    torch.manual_seed(2)
    M, N1, N2 = 256, 512, 128
    dy = torch.randn(M, N2, device=DEV).bfloat16()
    w2 = (torch.randn(N2, N1, device=DEV) * 0.05).bfloat16()
    pre = torch.randn(M, N1, device=DEV).bfloat16()
    dpre, db1 = ops.blaslt_gemm_dgelu_bgrad(dy, w2, pre)
    p = pre.float().requires_grad_(True)
    h = torch.nn.functional.gelu(p, approximate="tanh")
    (h * (dy.float() @ w2.float())).sum().backward()
    _close(dpre, p.grad, 0.3, "dgelu dgrad")
    _close(db1, p.grad.sum(dim=0), p.grad.abs().sum(0).max().item() * 2e-2 + 0.5, "bgrad")


@needs_aux_epilogues
def test_blaslt_mlp_autograd():
    torch.manual_seed(3)
    from dinov3_amd.ops.blaslt_mlp import blaslt_mlp

    M, D, H = 128, 64, 256
    x = torch.randn(M, D, device=DEV).bfloat16().requires_grad_(True)
    w1 = ((torch.randn(H, D, device=DEV)) * 0.05).bfloat16().requires_grad_(True)
    b1 = torch.randn(H, device=DEV).bfloat16().requires_grad_(True)
    w2 = ((torch.randn(D, H, device=DEV)) * 0.05).bfloat16().requires_grad_(True)
    b2 = torch.randn(D, device=DEV).bfloat16().requires_grad_(True)
    y = blaslt_mlp(x, w1, b1, w2, b2)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().requires_grad_(True)
    w1r = w1.detach().float().requires_grad_(True)
    b1r = b1.detach().float().requires_grad_(True)
    w2r = w2.detach().float().requires_grad_(True)
    b2r = b2.detach().float().requires_grad_(True)
    yr = torch.nn.functional.linear(
        torch.nn.functional.gelu(torch.nn.functional.linear(xr, w1r, b1r),
                                 approximate="tanh"), w2r, b2r)
    yr.backward(dy.float())
    _close(y, yr, 0.3, "mlp fwd")
    _close(x.grad, xr.grad, 0.3, "mlp dx")
    _close(w1.grad, w1r.grad, w1r.grad.abs().max().item() * 5e-2 + 0.3, "mlp dw1")
    _close(b1.grad, b1r.grad, b1r.grad.abs().max().item() * 5e-2 + 0.5, "mlp db1")
    _close(w2.grad, w2r.grad, w2r.grad.abs().max().item() * 5e-2 + 0.3, "mlp dw2")
