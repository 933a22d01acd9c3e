import numpy as np
import torch

from dinov3_amd.data import (
    DataAugmentationDINO,
    EpochSampler,
    InfiniteSampler,
    MaskingGenerator,
    ShardedInfiniteSampler,
    collate_data_and_cast,
    get_batch_subset,
    make_dataset,
)


def test_masking_generator_exact_count():
    gen = MaskingGenerator(input_size=(14, 14), max_num_patches=98)
    for n in (0, 10, 50, 98):
        mask = gen(n)
        assert mask.shape == (14, 14)
        assert int(mask.sum()) == n


def test_augmentation_output_shapes():
    aug = DataAugmentationDINO(
        global_crops_scale=(0.32, 1.0), local_crops_scale=(0.05, 0.32),
        local_crops_number=4, global_crops_size=64, local_crops_size=32,
    )
    img = torch.rand(3, 96, 96)
    out = aug(img)
    assert len(out["global_crops"]) == 2
    assert out["global_crops"][0].shape == (3, 64, 64)
    assert len(out["local_crops"]) == 4
    assert out["local_crops"][0].shape == (3, 32, 32)


def test_collate_crop_major_and_masks():
    aug_out = []
    B = 4
    for i in range(B):
        aug_out.append((
            {"global_crops": [torch.full((3, 8, 8), float(i)), torch.full((3, 8, 8), 10.0 + i)],
             "local_crops": [torch.full((3, 4, 4), 100.0 + i) for _ in range(3)]},
            (),
        ))
    gen = MaskingGenerator(input_size=(4, 4), max_num_patches=8)
    out = collate_data_and_cast(
        aug_out, mask_ratio_tuple=(0.1, 0.5), mask_probability=0.5, dtype=torch.float32,
        n_tokens=16, mask_generator=gen,
    )
    g = out["collated_global_crops"]
    assert g.shape == (2 * B, 3, 8, 8)
    # crop-major: first B entries are crop 0 of samples 0..B-1
    for i in range(B):
        assert g[i, 0, 0, 0].item() == float(i)
        assert g[B + i, 0, 0, 0].item() == 10.0 + i
    assert out["collated_masks"].shape == (2 * B, 16)
    n_masked = int(out["collated_masks"].sum())
    assert out["n_masked_patches"].item() == n_masked
    assert out["mask_indices_list"].numel() == n_masked
    assert out["masks_weight"].numel() == n_masked
    # per-sample mask weights sum to 1 for masked samples
    mw = out["masks_weight"]
    assert torch.all(mw > 0)


def test_get_batch_subset():
    B = 8
    batch = {
        "collated_global_crops": torch.randn(2 * B, 3, 8, 8),
        "collated_local_crops": torch.randn(4 * B, 3, 4, 4),
        "collated_masks": torch.zeros(2 * B, 16, dtype=torch.bool),
        "upperbound": 10,
    }
    batch["collated_masks"][0, :4] = True
    sub = get_batch_subset(batch, divide_by=2)
    assert sub["collated_global_crops"].shape[0] == B
    assert sub["collated_local_crops"].shape[0] == 2 * B


def test_epoch_sampler_rank_striding():
    s0 = EpochSampler(size=10, sample_count=10, shuffle=False, start=0, step=2)
    s1 = EpochSampler(size=10, sample_count=10, shuffle=False, start=1, step=2)
    i0, i1 = list(s0), list(s1)
    assert sorted(i0 + i1) == list(range(10))
    assert len(set(i0) & set(i1)) == 0


def test_epoch_sampler_shuffle_per_epoch():
    s = EpochSampler(size=16, sample_count=16, shuffle=True, seed=0, start=0, step=1)
    s.set_epoch(0)
    e0 = list(s)
    s.set_epoch(1)
    e1 = list(s)
    assert e0 != e1 and sorted(e0) == sorted(e1)


def test_infinite_samplers():
    import itertools

    inf = InfiniteSampler(sample_count=6, shuffle=True, seed=0, start=0, step=2)
    got = list(itertools.islice(iter(inf), 9))
    assert len(got) == 9
    sharded = ShardedInfiniteSampler(sample_count=6, shuffle=True, seed=0, start=1, step=2)
    got2 = list(itertools.islice(iter(sharded), 6))
    assert all(0 <= i < 6 for i in got2)


def test_make_dataset_string_parsing():
    ds = make_dataset(dataset_str="Synthetic:split=TRAIN:length=32")
    assert len(ds) == 32
    img, target = ds[0]
    assert img.shape == (3, 224, 224)
    ds2 = make_dataset(dataset_str="ImageNet:split=VAL")
    assert len(ds2) == 50_000


class TestGpuAugment:
    def test_batched_jitter_matches_per_image(self):
        import random
        from dinov3_amd.data import transforms as T
        from dinov3_amd.data.gpu_augment import (
            batched_brightness, batched_contrast, batched_hue, batched_saturation)

        torch.manual_seed(0)
        imgs = torch.rand(4, 3, 32, 32)
        bf = torch.tensor([0.7, 1.0, 1.3, 1.1])
        cf = torch.tensor([0.8, 1.2, 1.0, 0.9])
        sf = torch.tensor([1.1, 0.9, 1.0, 1.2])
        hf = torch.tensor([0.05, -0.08, 0.0, 0.1])
        outs = {
            "b": batched_brightness(imgs, bf),
            "c": batched_contrast(imgs, cf),
            "s": batched_saturation(imgs, sf),
            "h": batched_hue(imgs, hf),
        }
        for i in range(4):
            assert torch.allclose(outs["b"][i], T.adjust_brightness(imgs[i], float(bf[i])), atol=1e-6)
            assert torch.allclose(outs["c"][i], T.adjust_contrast(imgs[i], float(cf[i])), atol=1e-5)
            assert torch.allclose(outs["s"][i], T.adjust_saturation(imgs[i], float(sf[i])), atol=1e-6)
            assert torch.allclose(outs["h"][i], T.adjust_hue(imgs[i], float(hf[i])), atol=1e-5)

    def test_batched_blur_matches_per_image(self):
        from dinov3_amd.data.gpu_augment import GAUSS_KSIZE, batched_gaussian_blur

        torch.manual_seed(1)
        imgs = torch.rand(3, 3, 24, 24)
        sigma = torch.tensor([0.5, 1.0, 1.4])
        apply = torch.tensor([True, True, False])
        out = batched_gaussian_blur(imgs, sigma, apply)
        # identity where not applied
        assert torch.allclose(out[2], imgs[2], atol=1e-6)
        # per-image reference with the same capped kernel
        for i in range(2):
            k = GAUSS_KSIZE
            x = torch.arange(k, dtype=torch.float32) - k // 2
            kern = torch.exp(-0.5 * (x / float(sigma[i])) ** 2)
            kern = kern / kern.sum()
            kx = kern.view(1, 1, 1, k).expand(3, 1, 1, k)
            ky = kern.view(1, 1, k, 1).expand(3, 1, k, 1)
            ref = torch.nn.functional.conv2d(imgs[i:i+1], kx, padding=(0, k // 2), groups=3)
            ref = torch.nn.functional.conv2d(ref, ky, padding=(k // 2, 0), groups=3)
            assert torch.allclose(out[i], ref[0], atol=1e-5)

    def test_batched_rrc_flip_box_semantics(self):
        from dinov3_amd.data.gpu_augment import batched_rrc_flip

        torch.manual_seed(2)
        img = torch.rand(1, 3, 64, 64)
        # full-image box, no flip -> bilinear resize of the whole image
        boxes = torch.tensor([[0.0, 0.0, 64.0, 64.0]])
        out = batched_rrc_flip(img, boxes, torch.tensor([False]), 32)
        ref = torch.nn.functional.interpolate(img, size=(32, 32), mode="bilinear",
                                              align_corners=False)
        assert (out - ref).abs().max() < 2e-2
        # flip equivariance: flip(crop(img)) == crop_flipped(img)
        out_f = batched_rrc_flip(img, boxes, torch.tensor([True]), 32)
        assert torch.allclose(out_f, out.flip(-1), atol=1e-5)

    def test_full_pipeline_shapes_and_stats(self):
        import random
        from dinov3_amd.data.gpu_augment import GpuDataAugmentationDINO

        random.seed(3)
        torch.manual_seed(3)
        aug = GpuDataAugmentationDINO(local_crops_number=4, global_crops_size=64,
                                      local_crops_size=32)
        imgs = (torch.rand(5, 3, 96, 96) * 255).to(torch.uint8)
        out = aug(imgs)
        assert out["global_crops"].shape == (10, 3, 64, 64)
        assert out["local_crops"].shape == (20, 3, 32, 32)
        assert torch.isfinite(out["global_crops"]).all()
        # normalized output: roughly zero-centred
        assert out["global_crops"].mean().abs() < 2.0


def test_gpu_augment_pipeline_contract_cpu():
    """GpuAugmentPipeline emits collate_data_and_cast-shaped batches (CPU run
    of the same code path the GPU uses)."""
    import types

    import torch

    from dinov3_amd.configs import get_default_config
    from dinov3_amd.data.gpu_pipeline import build_gpu_augment_pipeline_from_cfg

    cfg = get_default_config()
    cfg.train.batch_size_per_gpu = 3
    cfg.train.num_workers = 0
    cfg.train.dataset_path = "Synthetic:split=TRAIN:length=16"
    cfg.crops.global_crops_size = 64
    cfg.crops.local_crops_size = 32
    cfg.crops.local_crops_number = 4
    pipe = build_gpu_augment_pipeline_from_cfg(cfg, torch.device("cpu"), torch.float32,
                                               canonical_size=96)
    batch = next(iter(pipe))
    B = 3
    assert batch["collated_global_crops"].shape == (2 * B, 3, 64, 64)
    assert batch["collated_local_crops"].shape == (4 * B, 3, 32, 32)
    n_tokens = (64 // cfg.student.patch_size) ** 2
    assert batch["collated_masks"].shape == (2 * B, n_tokens)
    assert batch["mask_indices_list"].numel() == int(batch["n_masked_patches"][0])
    assert batch["masks_weight"].numel() == batch["mask_indices_list"].numel()
    # normalized outputs: roughly zero-mean after ImageNet normalization
    assert batch["collated_global_crops"].abs().mean() < 5.0


def test_canonical_decode_sizes():
    import torch

    from dinov3_amd.data.gpu_pipeline import CanonicalDecode

    dec = CanonicalDecode(64)
    for h, w in ((100, 80), (64, 64), (50, 200)):
        out = dec(torch.rand(3, h, w))
        assert out.shape == (3, 64, 64)
        assert out.dtype == torch.uint8


def test_transform_presets():
    import torch

    from dinov3_amd.data.transforms import (
        get_target_transform,
        imaterialist_classification_target_transform,
        make_classification_eval_transform,
        make_classification_train_transform,
        make_eval_transform,
        voc2007_classification_target_transform,
    )

    img = torch.rand(3, 100, 80)
    assert make_classification_eval_transform()(img).shape == (3, 224, 224)
    assert make_classification_train_transform(crop_size=64)(img).shape == (3, 64, 64)
    assert make_eval_transform(resize_size=64, crop_size=0,
                               resize_square=True)(img).shape == (3, 64, 64)
    big = make_eval_transform(resize_size=64, crop_size=0, resize_large_side=True)(img)
    assert max(big.shape[1:]) == 64

    class _Inst:
        def __init__(self, cid):
            self.category_id = cid

    class _Label:
        instances = [_Inst(3), _Inst(7)]

    one_hot = voc2007_classification_target_transform(_Label())
    assert one_hot.sum() == 2 and one_hot[3] == 1 and one_hot[7] == 1

    class _IMat:
        attributes = [1, 5]

    assert imaterialist_classification_target_transform(_IMat()).sum() == 2
    assert get_target_transform("VOC2007:split=TRAIN") is voc2007_classification_target_transform
    assert get_target_transform("ImageNet:split=TRAIN") is None


def test_augmentation_option_matrix():
    """Exercise the augmentation options the trainer can reach from config:
    local-crops-as-subwindows, shared color jitter, no flips, gram crops
    without distortions."""
    import torch

    from dinov3_amd.data import DataAugmentationDINO

    img = torch.rand(3, 96, 96)
    for kwargs in (
        dict(local_crops_subset_of_global_crops=True, patch_size=16),
        dict(share_color_jitter=True),
        dict(horizontal_flips=False),
        dict(gram_teacher_crops_size=64, gram_teacher_no_distortions=True),
    ):
        aug = DataAugmentationDINO((0.32, 1.0), (0.05, 0.32), 3,
                                   global_crops_size=64, local_crops_size=32, **kwargs)
        out = aug(img)
        assert len(out["global_crops"]) == 2
        assert all(c.shape == (3, 64, 64) for c in out["global_crops"])
        assert len(out["local_crops"]) == 3
        assert all(c.shape == (3, 32, 32) for c in out["local_crops"])
        assert all(torch.isfinite(c).all() for c in out["global_crops"] + out["local_crops"])
        if "gram_teacher_crops_size" in kwargs:
            assert len(out["gram_teacher_crops"]) == 2
            assert out["gram_teacher_crops"][0].shape == (3, 64, 64)
