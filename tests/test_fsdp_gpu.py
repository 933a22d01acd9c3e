"""ShardedEngine GPU path (the code the driver's 2/4/8-GPU scale run uses):
world-size-1 RCCL process group exercises reduce_scatter_tensor /
all_gather_into_tensor plus the planned fp32-grad AdamW and fp32 l2norm
kernels, checked against plain fp32 math."""

import os

import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _dist_teardown():
    yield
    if dist.is_initialized():
        dist.destroy_process_group()


def _init_world1():
    if dist.is_initialized():
        return
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29631")
    dist.init_process_group("nccl", rank=0, world_size=1)


def _make_model():
    torch.manual_seed(5)
    m = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.GELU(), torch.nn.Linear(128, 64),
        torch.nn.LayerNorm(64),
    )
    return m.cuda().bfloat16()


def _make_groups(model):
    decay, no_decay = [], []
    for name, p in model.named_parameters():
        (no_decay if (name.endswith("bias") or "3." in name) else decay).append(p)
    return [
        {"params": decay, "names": [f"d{i}" for i in range(len(decay))],
         "submodel": "backbone", "lr_multiplier": 1.0, "wd_multiplier": 1.0,
         "is_last_layer": False},
        {"params": no_decay, "names": [f"n{i}" for i in range(len(no_decay))],
         "submodel": "backbone", "lr_multiplier": 0.5, "wd_multiplier": 0.0,
         "is_last_layer": False},
    ]


def test_sharded_engine_gpu_world1():
    _init_world1()
    from dinov3_amd.parallel.fsdp import ShardedEngine

    lr, wd, clip = 0.05, 0.1, 1.0
    model = _make_model()
    groups = _make_groups(model)
    # fp32 reference state captured BEFORE the engine re-points p.data
    ref = {id(p): {"w": p.detach().float().clone(), "m": torch.zeros_like(p, dtype=torch.float32),
                   "v": torch.zeros_like(p, dtype=torch.float32)}
           for g in groups for p in g["params"]}

    engine = ShardedEngine(groups, align=64)
    grads = {}
    for step in range(1, 3):
        torch.manual_seed(100 + step)
        x = torch.randn(16, 64, device="cuda").bfloat16()
        loss = (model(x).float() ** 2).mean()
        loss.backward()
        engine.finalize_backward()
        sums = engine.grad_norm_sums()
        dist.all_reduce(sums)
        # engine grads are fp32 reduce-scattered copies of the bf16 .grad
        for g in groups:
            for p in g["params"]:
                grads[id(p)] = p.grad.detach().float().clone()
        engine.step(lr=lr, weight_decay=wd, clip_scales=engine.clip_factors(sums, clip))
        engine.zero_grad()

        # fp32 reference AdamW on the captured grads
        ref_sum = sum((g_ ** 2).sum() for g_ in grads.values())
        assert abs(float(sums.sum() - ref_sum)) / max(float(ref_sum), 1e-6) < 2e-2, \
            "planned fp32 l2norm disagrees with torch"
        scale = min(clip / (float(ref_sum) ** 0.5 + 1e-6), 1.0)
        bc1, bc2 = 1 - 0.9 ** step, 1 - 0.999 ** step
        for g in groups:
            glr = lr * g["lr_multiplier"]
            gwd = wd * g["wd_multiplier"]
            for p in g["params"]:
                st = ref[id(p)]
                gr = grads[id(p)] * scale
                st["m"].mul_(0.9).add_(gr, alpha=0.1)
                st["v"].mul_(0.999).addcmul_(gr, gr, value=0.001)
                st["w"].mul_(1.0 - glr * gwd)
                st["w"].add_((st["m"] / bc1) / ((st["v"] / bc2).sqrt() + 1e-8), alpha=-glr)

    for g in groups:
        for p in g["params"]:
            want = ref[id(p)]["w"]
            err = (p.detach().float() - want).abs().max().item()
            assert err < 0.02, f"sharded GPU step mismatch: {err}"


def test_sharded_engine_gpu_full_meta_arch_step():
    """Tiny full DINOv3 step through ShardedEngine on RCCL world 1 — the exact
    engine selection and kernel path an 8-GPU bench rank runs."""
    _init_world1()
    import types

    from dinov3_amd.configs import setup_config
    from dinov3_amd.parallel.fsdp import ShardedEngine
    from dinov3_amd.train.ssl_meta_arch import SSLMetaArch

    args = types.SimpleNamespace(
        config_file="dinov3_amd/configs/train/vits_smoke.yaml", opts=[], output_dir="")
    cfg = setup_config(args, apply_scaling=False)
    torch.manual_seed(7)
    model = SSLMetaArch(cfg).cuda().bfloat16()
    model.train()
    engine = ShardedEngine(model.get_params_groups(), align=64)

    import sys
    sys.path.insert(0, os.getcwd())
    from bench import make_synthetic_batch

    batch = make_synthetic_batch(cfg, torch.device("cuda"), torch.bfloat16, n_batches=1)[0]
    for it in range(2):
        loss, _ = model(batch, teacher_temp=0.07, iteration=it)
        assert torch.isfinite(loss), f"non-finite loss at iter {it}"
        loss.backward()
        engine.finalize_backward()
        sums = engine.grad_norm_sums()
        dist.all_reduce(sums)
        engine.step(lr=1e-3, weight_decay=0.01,
                    clip_scales=engine.clip_factors(sums, 3.0))
        engine.zero_grad()
        model.update_ema(0.9)
    for p in model.student_backbone.parameters():
        assert torch.isfinite(p).all()
