import copy

import pytest
import torch

from dinov3_amd.train.ssl_meta_arch import SSLMetaArch


def _synthetic_batch(cfg, device=torch.device("cpu"), dtype=torch.float32):
    import sys, os

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from bench import make_synthetic_batch

    return make_synthetic_batch(cfg, device, dtype, n_batches=1)[0]


def test_meta_arch_forward_backward(smoke_cfg):
    torch.manual_seed(0)
    model = SSLMetaArch(smoke_cfg)
    model.train()
    batch = _synthetic_batch(smoke_cfg)
    loss, metrics = model(batch, teacher_temp=0.07, iteration=0)
    assert torch.isfinite(loss)
    for key in ("dino_local_crops_loss", "dino_global_crops_loss", "koleo_loss", "ibot_loss"):
        assert key in metrics
    loss.backward()
    grads = [p.grad for p in model.student_backbone.parameters() if p.grad is not None]
    assert grads, "student got no gradients"
    # teacher must have no grads (no_grad teacher forward)
    assert all(p.grad is None for p in model.teacher_backbone.parameters())


def test_teacher_starts_as_student_copy(smoke_cfg):
    model = SSLMetaArch(smoke_cfg)
    for (tn, tp), (sn, sp) in zip(
        model.teacher_backbone.named_parameters(), model.student_backbone.named_parameters()
    ):
        assert tn == sn
        assert torch.equal(tp, sp)


def test_ema_updates_live_teacher(smoke_cfg):
    torch.manual_seed(0)
    model = SSLMetaArch(smoke_cfg)
    with torch.no_grad():
        for p in model.student_backbone.parameters():
            p.add_(1.0)
    t_before = next(model.teacher_backbone.parameters()).detach().clone()
    model.update_ema(momentum=0.5)
    t_after = next(model.teacher_backbone.parameters())
    s = next(model.student_backbone.parameters())
    assert torch.allclose(t_after, 0.5 * t_before + 0.5 * s, atol=1e-6)


def test_loss_weights_scale(smoke_cfg):
    """dino global/local scales follow the crop-pair counts."""
    model = SSLMetaArch(smoke_cfg)
    n_g, n_l = 2, smoke_cfg.crops.local_crops_number
    g_terms = n_g * (n_g - 1)
    l_terms = n_g * n_l
    assert model.dino_global_ignore_diagonal
    assert abs(g_terms / (g_terms + l_terms) + l_terms / (g_terms + l_terms) - 1.0) < 1e-9


def test_gram_loss_path(smoke_cfg):
    cfg = copy.deepcopy(smoke_cfg)
    cfg.gram.use_loss = True
    cfg.gram.ckpt = "ignore"
    cfg.gram.img_level = True
    cfg.gram.remove_neg = True
    torch.manual_seed(0)
    model = SSLMetaArch(cfg)
    model.train()
    batch = _synthetic_batch(cfg)
    loss, metrics = model(batch, teacher_temp=0.07, iteration=0)
    assert "gram_loss" in metrics
    assert torch.isfinite(loss)
    loss.backward()


def test_distillation_mode(smoke_cfg, tmp_path):
    """distillation.enabled: frozen teacher built from the distillation
    config, EMA disabled, checkpoint weights loaded into the teacher."""
    import copy
    import torch as _torch

    from dinov3_amd.train.ssl_meta_arch import SSLMetaArch

    # source run: a normal smoke model whose weights we distill from
    torch.manual_seed(0)
    src = SSLMetaArch(copy.deepcopy(smoke_cfg))
    ckpt = tmp_path / "teacher.pth"
    _torch.save({"model": src.state_dict()}, ckpt)

    cfg = copy.deepcopy(smoke_cfg)
    cfg.distillation.enabled = True
    cfg.distillation.full_cfg_path = "dinov3_amd/configs/train/vits_smoke.yaml"
    cfg.distillation.checkpoint_path = str(ckpt)
    torch.manual_seed(1)
    model = SSLMetaArch(cfg)
    assert model.is_distillation_enabled
    # teacher came from the checkpoint, not from this student
    t = dict(model.teacher_backbone.named_parameters())
    s = dict(model.student_backbone.named_parameters())
    src_t = dict(src.teacher_backbone.named_parameters())
    name = next(iter(t))
    assert torch.equal(t[name], src_t[name])
    assert not torch.equal(t[name], s[name])
    # EMA is a no-op
    before = t[name].clone()
    model.update_ema(0.5)
    assert torch.equal(dict(model.teacher_backbone.named_parameters())[name], before)
    # teacher takes no grad
    assert not any(p.requires_grad for p in model.teacher_backbone.parameters())


def test_gram_tokens_used_and_schedules(smoke_cfg):
    """gram.tokens_used masked/unmasked subsets and the gram/dino-local
    loss-weight schedules (options the reference validates but drops)."""
    cfg = copy.deepcopy(smoke_cfg)
    cfg.gram.use_loss = True
    cfg.gram.ckpt = "ignore"
    cfg.gram.img_level = False
    cfg.gram.remove_neg = True
    cfg.gram.tokens_used = "masked"
    cfg.gram.loss_weight_schedule = {"start": 0.0, "peak": 0.0, "end": 2.0,
                                     "warmup_epochs": 0, "cosine_epochs": 1}
    cfg.dino.reweight_dino_local_loss = True
    cfg.dino.local_loss_weight_schedule = {"start": 1.0, "peak": 1.0, "end": 0.5,
                                           "warmup_epochs": 0, "cosine_epochs": 1}
    torch.manual_seed(0)
    model = SSLMetaArch(cfg)
    model.train()
    batch = _synthetic_batch(cfg)
    loss0, m0 = model(batch, teacher_temp=0.07, iteration=0)
    assert torch.isfinite(loss0)
    total = cfg.train.OFFICIAL_EPOCH_LENGTH * cfg.optim.epochs
    loss1, m1 = model(batch, teacher_temp=0.07, iteration=total - 1)
    # schedules move over the run: gram weight 0 -> 2, local weight 1 -> 0.5
    assert abs(float(m0["gram_loss_weight"]) - 0.0) < 1e-6
    assert abs(float(m1["gram_loss_weight"]) - 2.0) < 1e-3
    assert abs(float(m0["dino_local_loss_weight"]) - 1.0) < 1e-6
    assert abs(float(m1["dino_local_loss_weight"]) - 0.5) < 1e-3


def test_gram_ema_teacher_mode(smoke_cfg):
    cfg = copy.deepcopy(smoke_cfg)
    cfg.gram.use_loss = True
    cfg.gram.ema_teacher = True
    cfg.gram.remove_neg = True
    cfg.gram.rep_update = False
    torch.manual_seed(0)
    model = SSLMetaArch(cfg)
    assert model.gram_backbone is None and not model.has_gram_teacher
    model.train()
    batch = _synthetic_batch(cfg)
    loss, metrics = model(batch, teacher_temp=0.07, iteration=0)
    assert "gram_loss" in metrics and torch.isfinite(loss)


def test_nan_check_sanitizer(monkeypatch):
    """DINOV3_NAN_CHECK=1 raises at the first block producing non-finites."""
    import importlib

    import torch

    import dinov3_amd.models.vision_transformer as vt

    monkeypatch.setenv("DINOV3_NAN_CHECK", "1")
    importlib.reload(vt)
    try:
        model = vt.vit_small(patch_size=16)
        model.eval()
        with torch.no_grad():
            # healthy input passes
            model([torch.randn(1, 3, 64, 64)], [None])
            # poison a weight -> the sanitizer must fire with the block index
            model.blocks[2].mlp.fc1.weight.data.fill_(float("inf"))
            with pytest.raises(FloatingPointError, match="after block 2"):
                model([torch.randn(1, 3, 64, 64)], [None])
    finally:
        monkeypatch.delenv("DINOV3_NAN_CHECK")
        importlib.reload(vt)
