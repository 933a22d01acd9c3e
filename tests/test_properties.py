"""Property-based tests (hypothesis) for invariant-heavy components."""

import hypothesis.strategies as st
import pytest
import torch
from hypothesis import given, settings


@settings(max_examples=40, deadline=None)
@given(h=st.integers(32, 96), w=st.integers(32, 96),
       lo=st.floats(0.05, 0.6), span=st.floats(0.05, 0.4))
def test_rrc_box_always_in_bounds(h, w, lo, span):
    from dinov3_amd.data.transforms import sample_rrc_box

    top, left, ch, cw = sample_rrc_box(h, w, (lo, min(1.0, lo + span)))
    assert 0 <= top and 0 <= left and ch > 0 and cw > 0
    assert top + ch <= h and left + cw <= w


@settings(max_examples=30, deadline=None)
@given(gh=st.integers(4, 14), gw=st.integers(4, 14), target=st.integers(0, 60))
def test_masking_generator_exact_count(gh, gw, target):
    from dinov3_amd.data.masking import MaskingGenerator

    target = min(target, gh * gw)
    gen = MaskingGenerator(input_size=(gh, gw), max_num_patches=max(gh * gw // 2, 1))
    mask = gen(target)
    assert mask.shape == (gh, gw)
    assert int(mask.sum()) == target


@settings(max_examples=25, deadline=None)
@given(b=st.integers(1, 17), keep_ratio=st.floats(0.1, 1.0), n=st.integers(1, 9),
       layers=st.integers(1, 6))
def test_droppath_plan_always_valid(b, keep_ratio, n, layers):
    from dinov3_amd.layers.block import DropPathPlan

    metas = [(0, b, n, None, None, 0)]
    plan = DropPathPlan(metas, keep_ratio, 2 * layers, torch.device("cpu"))
    keep = max(int(b * keep_ratio), 1)
    for s in range(2 * layers):
        rows, new_metas, scale = plan.take(s)
        assert rows.shape[0] == keep * n
        assert rows.min() >= 0 and rows.max() < b * n
        assert len(set(rows.tolist())) == rows.shape[0]
        assert new_metas[0][1] == keep
        assert torch.allclose(scale, torch.full_like(scale, b / keep))


@settings(max_examples=25, deadline=None)
@given(start=st.floats(0.0, 1.0), peak=st.floats(0.0, 2.0), end=st.floats(0.0, 2.0),
       warm=st.integers(0, 20), total=st.integers(21, 80))
def test_linear_warmup_cosine_decay_bounds(start, peak, end, warm, total):
    from dinov3_amd.train.cosine_lr_scheduler import linear_warmup_cosine_decay

    s = linear_warmup_cosine_decay(start=start, peak=peak, end=end,
                                   warmup_iterations=warm, total_iterations=total)
    lo = min(start, peak, end) - 1e-9
    hi = max(start, peak, end) + 1e-9
    for it in (0, warm, total - 1, total + 5):
        assert lo <= s[it] <= hi
    if warm > 0:
        assert abs(s[0] - start) < 1e-9
    assert abs(s[total + 100] - end) < 1e-9


@settings(max_examples=20, deadline=None)
@given(m=st.integers(2, 24), k=st.integers(3, 64), temp=st.floats(0.1, 2.0),
       iters=st.integers(1, 5))
def test_sinkhorn_rows_are_distributions(m, k, temp, iters):
    from dinov3_amd.loss.dino_clstoken_loss import sinkhorn_knopp

    x = torch.randn(m, k)
    probs = sinkhorn_knopp(x, temp, n_iterations=iters)
    assert probs.shape == (m, k)
    assert (probs >= 0).all()
    assert torch.allclose(probs.sum(dim=-1), torch.ones(m), atol=1e-4)


@settings(max_examples=25, deadline=None)
@given(b=st.integers(1, 6), nloc=st.integers(1, 4),
       prob=st.floats(0.0, 1.0), rmin=st.floats(0.05, 0.3), rspan=st.floats(0.05, 0.3))
def test_collate_mask_invariants(b, nloc, prob, rmin, rspan):
    from dinov3_amd.data import MaskingGenerator, collate_data_and_cast

    gs, ls, p = 32, 16, 16
    n_tokens = (gs // p) ** 2
    gen = MaskingGenerator(input_size=(gs // p, gs // p),
                           max_num_patches=max(1, n_tokens // 2))
    samples = [({"global_crops": [torch.randn(3, gs, gs) for _ in range(2)],
                 "local_crops": [torch.randn(3, ls, ls) for _ in range(nloc)]}, ())
               for _ in range(b)]
    out = collate_data_and_cast(samples, mask_ratio_tuple=(rmin, min(0.5, rmin + rspan)),
                                mask_probability=prob, dtype=torch.float32,
                                n_tokens=n_tokens, mask_generator=gen)
    masks = out["collated_masks"]
    idx = out["mask_indices_list"]
    w = out["masks_weight"]
    assert masks.shape == (2 * b, n_tokens)
    assert out["collated_global_crops"].shape == (2 * b, 3, gs, gs)
    assert out["collated_local_crops"].shape == (nloc * b, 3, ls, ls)
    # mask_indices_list is exactly the nonzero positions of the flat mask
    assert torch.equal(idx, masks.flatten().nonzero().flatten())
    assert int(out["n_masked_patches"][0]) == int(masks.sum())
    assert out["upperbound"] >= int(masks.sum())
    # per masked sample, its weights sum to 1
    assert w.shape[0] == idx.shape[0]
    if idx.numel():
        sample_of = idx // n_tokens
        for s in sample_of.unique().tolist():
            assert abs(float(w[sample_of == s].sum()) - 1.0) < 1e-5


@settings(max_examples=20, deadline=None)
@given(st.dictionaries(st.sampled_from(["train", "optim", "dino"]),
                       st.dictionaries(st.sampled_from(["a_key", "b_key"]),
                                       st.one_of(st.integers(), st.floats(allow_nan=False),
                                                 st.text(max_size=5)), max_size=2),
                       max_size=3))
def test_config_merge_roundtrip(over):
    """Merging a delta then re-diffing yields the same delta (tolerant mode)."""
    from dinov3_amd.configs import get_default_config
    from dinov3_amd.configs.config import _merge_into

    cfg = get_default_config()
    _merge_into(cfg, over, strict=False)
    for section, kv in over.items():
        for k, v in kv.items():
            got = cfg[section][k]
            assert got == v or (got != got and v != v)  # NaN-safe


def test_droppath_plan_uniform_inclusion():
    """argsort-of-uniforms subsets are uniform: every sample is kept with
    probability keep/B (chi-square-ish bound over many draws)."""
    from dinov3_amd.layers.block import DropPathPlan

    torch.manual_seed(123)
    B, keep_ratio, trials = 8, 0.5, 400
    counts = torch.zeros(B)
    metas = [(0, B, 1, None, None, 0)]
    for _ in range(trials):
        plan = DropPathPlan(metas, keep_ratio, 1, torch.device("cpu"))
        rows, _, _ = plan.take(0)
        counts[rows] += 1
    expected = trials * 0.5
    # std of a binomial(400, .5) is 10; allow 4 sigma
    assert ((counts - expected).abs() < 40).all(), counts
