import types

import pytest

from dinov3_amd.configs import get_default_config, setup_config
from dinov3_amd.configs.config import apply_dotlist


def test_default_schema_keys():
    cfg = get_default_config()
    for section in ("MODEL", "compute_precision", "dino", "ibot", "gram", "train",
                    "student", "teacher", "distillation", "multidistillation", "hrft",
                    "optim", "crops", "evaluation", "checkpointing"):
        assert section in cfg, section
    assert cfg.dino.head_n_prototypes == 65536
    assert cfg.train.centering == "sinkhorn_knopp"
    assert cfg.student.arch == "vit_large"
    assert cfg.compute_precision.sharding_strategy == "SHARD_GRAD_OP"


def test_dotlist_and_types():
    cfg = get_default_config()
    apply_dotlist(cfg, ["optim.lr=0.005", "student.arch=vit_small", "train.batch_size_per_gpu=8"])
    assert cfg.optim.lr == 0.005
    assert cfg.student.arch == "vit_small"
    assert cfg.train.batch_size_per_gpu == 8


def test_dotlist_unknown_key_rejected():
    cfg = get_default_config()
    with pytest.raises(KeyError):
        apply_dotlist(cfg, ["optim.not_a_key=1"])


def test_scaling_rule_sqrt():
    args = types.SimpleNamespace(config_file="", opts=["train.batch_size_per_gpu=1024"], output_dir="")
    cfg = setup_config(args, apply_scaling=True)
    # sqrt_wrt_1024 at world=1, batch 1024 -> lr unchanged
    assert abs(cfg.optim.lr - 0.001) < 1e-9


def test_reference_style_config_loads(smoke_cfg):
    assert smoke_cfg.student.arch == "vit_small"
    assert smoke_cfg.dino.head_n_prototypes == 64
