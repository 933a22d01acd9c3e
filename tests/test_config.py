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


def test_reference_recipe_configs_load():
    """The reference's named train recipes exist here as delta configs and
    produce complete configs through the merge (unknown recipe-only keys are
    tolerated by the file merge)."""
    import types

    from dinov3_amd.configs import setup_config

    for name in [
        "dinov3_vit7b16_pretrain",
        "dinov3_vit7b16_gram_anchor",
        "dinov3_vit7b16_high_res_adapt",
        "vitl_im1k_lin834_smol",
        "multi_distillation_test",
    ]:
        args = types.SimpleNamespace(
            config_file=f"dinov3_amd/configs/train/{name}.yaml", opts=[], output_dir="")
        cfg = setup_config(args, apply_scaling=False)
        assert cfg.train.batch_size_per_gpu > 0
        assert cfg.student.arch.startswith("vit") or cfg.multidistillation.enabled
    # vit7b pretrain recipe specifics survive the merge
    args = types.SimpleNamespace(
        config_file="dinov3_amd/configs/train/dinov3_vit7b16_pretrain.yaml",
        opts=[], output_dir="")
    cfg = setup_config(args, apply_scaling=False)
    assert cfg.student.arch == "vit_7b"
    assert cfg.dino.head_n_prototypes == 262144
    assert cfg.student.ffn_layer == "swiglu64"
    assert cfg.crops.global_crops_size == 256
