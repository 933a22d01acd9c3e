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
