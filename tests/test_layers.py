import math

import pytest
import torch

from dinov3_amd.layers import (
    DINOHead,
    LayerNorm,
    Mlp,
    PatchEmbed,
    RMSNorm,
    RopePositionEmbedding,
    SelfAttention,
    SelfAttentionBlock,
    SwiGLUFFN,
)
from dinov3_amd.models.vision_transformer import vit_small


def test_patch_embed_matches_conv():
    torch.manual_seed(0)
    pe = PatchEmbed(img_size=32, patch_size=16, embed_dim=64)
    x = torch.randn(2, 3, 32, 32)
    out, hp, wp = pe(x)
    assert out.shape == (2, 4, 64) and hp == wp == 2
    # equivalence with a stride-p conv carrying the same weights
    conv = torch.nn.Conv2d(3, 64, 16, stride=16)
    # conv-native (c, dy, dx) row flattening: conv weight is a plain reshape
    conv.weight.data.copy_(pe.proj.weight.reshape(64, 3, 16, 16))
    conv.bias.data.copy_(pe.proj.bias)
    ref = conv(x).flatten(2).transpose(1, 2)
    assert torch.allclose(out, ref, atol=1e-4)


def test_rope_tables_shape_and_range():
    rope = RopePositionEmbedding(embed_dim=384, num_heads=6)
    sin, cos = rope(H=14, W=14)
    assert sin.shape == (196, 64) and cos.shape == (196, 64)
    assert torch.all(sin.abs() <= 1.0001) and torch.all(cos.abs() <= 1.0001)
    # duplicated halves: angles[:, :32] == angles[:, 32:]
    assert torch.allclose(sin[:, :32], sin[:, 32:])


def test_attention_shapes_and_rope_prefix():
    torch.manual_seed(0)
    attn = SelfAttention(dim=64, num_heads=4, qkv_bias=True)
    x = torch.randn(2, 17, 64)  # 1 cls + 16 patches
    rope = RopePositionEmbedding(embed_dim=64, num_heads=4)
    sincos = rope(H=4, W=4)
    out = attn(x, rope=sincos)
    assert out.shape == x.shape


def test_block_list_forward_matches_single():
    torch.manual_seed(0)
    blk = SelfAttentionBlock(dim=64, num_heads=4, qkv_bias=True, init_values=1e-5)
    blk.eval()
    xg = torch.randn(2, 17, 64)
    xl = torch.randn(4, 5, 64)
    outs = blk.forward_list([xg, xl], [None, None])
    single_g = blk(xg)
    assert torch.allclose(outs[0], single_g, atol=1e-5)


def test_droppath_subset_training():
    torch.manual_seed(0)
    blk = SelfAttentionBlock(dim=32, num_heads=2, drop_path=0.5, init_values=1e-5)
    blk.train()
    x = torch.randn(8, 5, 32)
    out = blk(x)
    assert out.shape == x.shape


def test_mlp_no_activation_after_fc2():
    torch.manual_seed(0)
    mlp = Mlp(in_features=16, hidden_features=32)
    x = torch.randn(4, 16)
    ref = mlp.fc2(torch.nn.functional.gelu(mlp.fc1(x), approximate="tanh"))
    out = mlp(x)
    assert torch.allclose(out, ref, atol=1e-5)
    # a gelu after fc2 would make outputs nonnegative-biased; check sign variety
    assert (out < 0).any()


def test_swiglu_hidden_alignment():
    ffn = SwiGLUFFN(in_features=64, hidden_features=256, align_to=64)
    assert ffn.hidden_features % 64 == 0
    out = ffn(torch.randn(3, 64))
    assert out.shape == (3, 64)


def test_dino_head_l2_bottleneck():
    torch.manual_seed(0)
    head = DINOHead(in_dim=32, out_dim=128, nlayers=3, hidden_dim=64, bottleneck_dim=16)
    x = torch.randn(10, 32)
    pre = head(x, no_last_layer=True)
    norms = pre.norm(dim=-1)
    assert torch.allclose(norms, torch.ones_like(norms), atol=1e-3)
    out = head(x)
    assert out.shape == (10, 128)


def test_vit_forward_features_list():
    torch.manual_seed(0)
    model = vit_small(img_size=64, drop_path_rate=0.0, layerscale_init=1e-5)
    model.eval()
    g = torch.randn(2, 3, 64, 64)
    l = torch.randn(4, 3, 32, 32)
    out = model.forward_features([g, l], [None, None])
    assert out[0]["x_norm_clstoken"].shape == (2, 384)
    assert out[0]["x_norm_patchtokens"].shape == (2, 16, 384)
    assert out[1]["x_norm_patchtokens"].shape == (4, 4, 384)


def test_vit_storage_tokens_and_masks():
    model = vit_small(img_size=32, n_storage_tokens=4, layerscale_init=1e-5)
    model.eval()
    x = torch.randn(2, 3, 32, 32)
    masks = torch.zeros(2, 4, dtype=torch.bool)
    masks[0, 1] = True
    out = model.forward_features(x, masks)
    assert out["x_storage_tokens"].shape == (2, 4, 384)
    assert out["x_norm_patchtokens"].shape == (2, 4, 384)


def test_get_intermediate_layers():
    model = vit_small(img_size=32, layerscale_init=1e-5)
    model.eval()
    x = torch.randn(1, 3, 32, 32)
    outs = model.get_intermediate_layers(x, n=2, return_class_token=True)
    assert len(outs) == 2
    patches, cls = outs[0]
    assert patches.shape == (1, 4, 384) and cls.shape == (1, 384)


def test_droppath_plan_mechanics():
    from dinov3_amd.layers.block import DropPathPlan

    torch.manual_seed(0)
    metas = [(0, 4, 5, None, None, 1), (20, 6, 3, None, None, 1)]
    plan = DropPathPlan(metas, keep_ratio=0.5, n_sublayers=6, device="cpu")
    assert plan.rows.shape[0] == 6
    for s in range(6):
        rows, new_metas, scale = plan.take(s)
        # group 0: keep 2 of 4 samples -> 10 rows in [0, 20); group 1: 3 of 6 -> 9 rows in [20, 38)
        assert rows.shape[0] == 2 * 5 + 3 * 3
        g0, g1 = rows[:10], rows[10:]
        assert g0.min() >= 0 and g0.max() < 20 and g1.min() >= 20 and g1.max() < 38
        assert len(set(rows.tolist())) == rows.shape[0]  # no duplicate rows
        assert torch.allclose(scale[:10], torch.full((10,), 2.0))
        assert torch.allclose(scale[10:], torch.full((9,), 2.0))
    # different sublayers draw different subsets (overwhelmingly likely)
    assert not torch.equal(plan.rows[0], plan.rows[1])


def test_droppath_plan_keepall_matches_dense():
    """keep == B makes drop-path a no-op: the planned subset path must equal
    the plain dense forward bit-for-bit (scatter-add is order-independent)."""
    from dinov3_amd.layers.attention import SelfAttention
    from dinov3_amd.layers.block import DropPathPlan
    from dinov3_amd.utils.utils import cat_keep_shapes

    torch.manual_seed(0)
    blk = SelfAttentionBlock(dim=32, num_heads=2, drop_path=0.5, init_values=1e-5)
    blk.train()
    x = torch.randn(6, 5, 32)
    flat, _, _ = cat_keep_shapes([x])
    metas = [SelfAttention._meta_for(x, None, 0)]
    plan = DropPathPlan(metas, keep_ratio=1.0, n_sublayers=2, device="cpu")
    out_plan = blk.forward_flat(flat.clone(), metas, plan, 0)
    blk.eval()
    out_dense = blk.forward_flat(flat.clone(), metas)
    assert torch.allclose(out_plan, out_dense, atol=1e-5)


def _block_run(fused, drop_path, monkeypatch, checkpoint=False):
    import os
    from dinov3_amd.layers.attention import SelfAttention
    from dinov3_amd.utils.utils import cat_keep_shapes
    from dinov3_amd.layers.block import DropPathPlan

    monkeypatch.setenv("DINOV3_FUSED_RESIDUAL", "1" if fused else "0")
    torch.manual_seed(11)
    blk = SelfAttentionBlock(dim=32, num_heads=2, qkv_bias=True,
                             drop_path=drop_path, init_values=1e-2)
    blk.train()
    torch.manual_seed(12)
    x = torch.randn(6, 5, 32)
    flat, _, _ = cat_keep_shapes([x])
    flat = flat.clone().requires_grad_(True)
    metas = [SelfAttention._meta_for(x, None, 0)]
    plan = None
    if drop_path > 0:
        torch.manual_seed(13)
        plan = DropPathPlan(metas, 1.0 - drop_path, 2, flat.device)
    out = blk.forward_flat(flat, metas, plan, 0, inplace_ok=not checkpoint)
    out.float().pow(2).sum().backward()
    grads = {n: p.grad.clone() for n, p in blk.named_parameters()}
    return out.detach().clone(), grads, flat.grad.clone()


@pytest.mark.parametrize("drop_path", [0.0, 0.5])
def test_fused_residual_matches_default(drop_path, monkeypatch):
    """DINOV3_FUSED_RESIDUAL folds gamma+proj/fc2-bias into the residual op;
    on CPU both paths are plain torch math and must agree to fp tolerance."""
    out_a, grads_a, dx_a = _block_run(False, drop_path, monkeypatch)
    out_b, grads_b, dx_b = _block_run(True, drop_path, monkeypatch)
    assert torch.allclose(out_a, out_b, atol=1e-5), (out_a - out_b).abs().max()
    assert torch.allclose(dx_a, dx_b, atol=1e-5)
    for n in grads_a:
        assert torch.allclose(grads_a[n], grads_b[n], atol=1e-4), \
            f"{n}: {(grads_a[n] - grads_b[n]).abs().max()}"


def test_patch14_forward_backward():
    """Non-16 patch sizes (e.g. the ViT-g/14 target config) take the
    patchify+GEMM fallback; full fwd+bwd must work."""
    from dinov3_amd.models.vision_transformer import vit_small

    torch.manual_seed(0)
    m = vit_small(img_size=56, patch_size=14, layerscale_init=1e-5)
    m.train()
    x = torch.randn(2, 3, 56, 56)
    out = m.forward_features(x)
    assert out["x_norm_patchtokens"].shape == (2, 16, 384)
    out["x_norm_clstoken"].float().pow(2).sum().backward()
    assert m.patch_embed.proj.weight.grad is not None
