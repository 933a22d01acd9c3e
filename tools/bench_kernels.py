"""Per-kernel micro-benchmarks: each HIP op vs its plain-torch counterpart.

GPU-only. Usage (on a GPU box):
    python tools/bench_kernels.py [--iters 50] [--shape-set vitl]

Prints one line per op: name, HIP us, torch us, speedup, effective GB/s of
the HIP path (bytes moved at the op's minimum traffic).
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def row(name, hip_us, ref_us, bytes_moved):
    gbs = bytes_moved / (hip_us * 1e-6) / 1e9 if hip_us > 0 else 0.0
    print(f"{name:34s} hip {hip_us:9.1f} us   torch {ref_us:9.1f} us   "
          f"x{ref_us / hip_us:5.2f}   {gbs:7.0f} GB/s")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=50)
    args = p.parse_args()
    assert torch.cuda.is_available(), "GPU required"
    dev = "cuda"
    it = args.iters

    # ViT-L flat-buffer shapes: R = 2*64*197 + 8*64*37 tokens, D = 1024
    R, D = 2 * 64 * 197 + 8 * 64 * 37, 1024
    H4 = 4 * D

    from dinov3_amd.ops import bias_gelu, hip_ops, l2_normalize, layer_norm
    from dinov3_amd.ops.ls_axpy import ls_axpy

    ops = hip_ops()
    x = torch.randn(R, D, device=dev).bfloat16()
    w = torch.ones(D, device=dev).bfloat16()
    b = torch.zeros(D, device=dev).bfloat16()

    row("layernorm_fwd [R,1024]",
        timeit(lambda: layer_norm(x, w, b), it),
        timeit(lambda: torch.nn.functional.layer_norm(x, (D,), w, b), it),
        R * D * 2 * 2)

    h = torch.randn(R, H4, device=dev).bfloat16()
    hb = torch.randn(H4, device=dev).bfloat16()
    row("bias_gelu_fwd [R,4096]",
        timeit(lambda: bias_gelu(h, hb), it),
        timeit(lambda: torch.nn.functional.gelu(h + hb, approximate="tanh"), it),
        R * H4 * 2 * 2)

    res = torch.randn_like(x)
    gamma = torch.full((D,), 1e-5, device=dev).bfloat16()
    row("ls_axpy_fwd [R,1024]",
        timeit(lambda: ls_axpy(x, res, gamma), it),
        timeit(lambda: x + gamma * res, it),
        R * D * 2 * 3)

    row("l2norm_fwd [8192,256]",
        timeit(lambda: l2_normalize(x[:8192, :256]), it),
        timeit(lambda: torch.nn.functional.normalize(x[:8192, :256].float(), dim=-1), it),
        8192 * 256 * 2 * 2)

    # FMHA fwd: global-crop group
    from dinov3_amd.ops.flat_attention import flat_multi_fmha

    B, N, Hh, hd = 128, 197, 16, 64
    qkv = torch.randn(B * N, 3 * Hh * hd, device=dev).bfloat16()
    Pn = N - 1
    a = torch.rand(Pn, hd // 2, device=dev)
    a = torch.cat([a, a], dim=-1)
    sin, cos = a.sin().contiguous(), a.cos().contiguous()
    metas = [(0, B, N, sin, cos, 1)]
    q = qkv.view(B, N, 3, Hh, hd)[:, :, 0].permute(0, 2, 1, 3)

    def sdpa():
        qq = q.contiguous()
        return torch.nn.functional.scaled_dot_product_attention(qq, qq, qq)

    row("fmha_rope_fwd [128,197,16,64]",
        timeit(lambda: flat_multi_fmha(qkv, Hh, metas), it),
        timeit(sdpa, it),
        B * N * 3 * Hh * hd * 2 * 2)

    # FMHA hd-128 at the high-res-adapt shape: ViT-7b/16 @768px -> N=2305
    # (dinov3_vit7b16_high_res_adapt.yaml:162-186), 32 heads x d128
    B2, N2, Hh2, hd2 = 4, 2305, 32, 128
    qkv2 = torch.randn(B2 * N2, 3 * Hh2 * hd2, device=dev).bfloat16()
    Pn2 = N2 - 1
    a2 = torch.rand(Pn2, hd2 // 2, device=dev)
    a2 = torch.cat([a2, a2], dim=-1)
    sin2, cos2 = a2.sin().contiguous(), a2.cos().contiguous()
    metas2 = [(0, B2, N2, sin2, cos2, 1)]
    q2 = qkv2.view(B2, N2, 3, Hh2, hd2)[:, :, 0].permute(0, 2, 1, 3)

    def sdpa2():
        qq = q2.contiguous()
        return torch.nn.functional.scaled_dot_product_attention(qq, qq, qq)

    row("fmha_rope_fwd [4,2305,32,128]",
        timeit(lambda: flat_multi_fmha(qkv2, Hh2, metas2), it),
        timeit(sdpa2, it),
        B2 * N2 * 3 * Hh2 * hd2 * 2 * 2)

    def fmha2_fwd_bwd():
        qkv_g = qkv2.detach().requires_grad_(True)
        out = flat_multi_fmha(qkv_g, Hh2, metas2)
        out.float().sum().backward()

    def sdpa2_fwd_bwd():
        qq = q2.detach().contiguous().requires_grad_(True)
        out = torch.nn.functional.scaled_dot_product_attention(qq, qq, qq)
        out.float().sum().backward()

    row("fmha_rope_f+b [4,2305,32,128]",
        timeit(fmha2_fwd_bwd, max(it // 4, 3)),
        timeit(sdpa2_fwd_bwd, max(it // 4, 3)),
        B2 * N2 * 3 * Hh2 * hd2 * 2 * 6)

    # fused sinkhorn over K=65536 (teacher shapes: M=128 cls rows)
    from dinov3_amd.ops.proto_scores import sinkhorn_knopp_factored

    K = 65536
    logits = torch.randn(128, K, device=dev).bfloat16()

    def dense_sinkhorn():
        Q = (logits.float() / 0.07).T.exp()
        Q = Q / Q.sum()
        for _ in range(3):
            Q = Q / Q.sum(dim=1, keepdim=True) / K
            Q = Q / Q.sum(dim=0, keepdim=True) / 128
        return Q

    row("sinkhorn K=65536 M=128 (3 it)",
        timeit(lambda: sinkhorn_knopp_factored(logits, 0.07, 3), it),
        timeit(dense_sinkhorn, it),
        128 * K * 2 * 6)

    # patch-embed GEMM
    xi = torch.randn(64, 3, 224, 224, device=dev).bfloat16()
    wpe = (torch.randn(1024, 768, device=dev) * 0.02).bfloat16()
    bpe = torch.zeros(1024, device=dev).bfloat16()
    conv_w = wpe.reshape(1024, 3, 16, 16)

    row("patch_embed [64,3,224,224]",
        timeit(lambda: ops.patch_embed_fwd(xi, wpe, bpe, 16), it),
        timeit(lambda: torch.nn.functional.conv2d(xi, conv_w, bpe, stride=16), it),
        64 * 3 * 224 * 224 * 2 + 64 * 196 * 1024 * 2)


if __name__ == "__main__":
    main()
