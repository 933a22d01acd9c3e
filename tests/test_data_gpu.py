"""On-device batched augmentation: full pipeline on the GPU (pure tensor
ops — validates the grid_sample / grouped-conv path on ROCm)."""

import random

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_gpu_augment_pipeline_on_device():
    from dinov3_amd.data.gpu_augment import GpuDataAugmentationDINO

    random.seed(0)
    torch.manual_seed(0)
    aug = GpuDataAugmentationDINO(local_crops_number=4)
    imgs = (torch.rand(8, 3, 256, 256, device="cuda") * 255).to(torch.uint8)
    out = aug(imgs)
    assert out["global_crops"].shape == (16, 3, 224, 224)
    assert out["local_crops"].shape == (32, 3, 96, 96)
    assert out["global_crops"].is_cuda and torch.isfinite(out["global_crops"]).all()
    assert torch.isfinite(out["local_crops"]).all()


def test_gpu_augment_matches_cpu():
    """Same host RNG draws -> identical crops up to fp tolerance on GPU."""
    from dinov3_amd.data.gpu_augment import GpuDataAugmentationDINO

    imgs = torch.rand(4, 3, 128, 128)
    aug = GpuDataAugmentationDINO(local_crops_number=2, global_crops_size=64,
                                  local_crops_size=32)
    random.seed(42)
    cpu = aug(imgs)
    random.seed(42)
    gpu = aug(imgs.cuda())
    for k in ("global_crops", "local_crops"):
        err = (cpu[k] - gpu[k].cpu()).abs().max().item()
        assert err < 5e-2, f"{k}: max err {err}"


def test_gpu_augment_training_pipeline_end_to_end():
    """GpuAugmentPipeline feeds a real train step on device: decode-only
    loader -> batched device augmentation -> collate-contract batch -> model."""
    from dinov3_amd.configs import get_default_config
    from dinov3_amd.data.gpu_pipeline import build_gpu_augment_pipeline_from_cfg

    cfg = get_default_config()
    cfg.train.batch_size_per_gpu = 4
    cfg.train.num_workers = 0
    cfg.train.dataset_path = "Synthetic:split=TRAIN:length=16"
    cfg.crops.local_crops_number = 4
    pipe = build_gpu_augment_pipeline_from_cfg(cfg, torch.device("cuda:0"), torch.bfloat16)
    batch = next(iter(pipe))
    assert batch["collated_global_crops"].is_cuda
    assert batch["collated_global_crops"].dtype == torch.bfloat16
    assert batch["collated_global_crops"].shape == (8, 3, 224, 224)
    assert batch["collated_local_crops"].shape == (16, 3, 96, 96)
    assert batch["mask_indices_list"].is_cuda
    assert int(batch["n_masked_patches"][0]) == batch["mask_indices_list"].numel()
