"""Multi-distillation: per-student rank subgroups with scoped collectives.
2 students x 2 ranks on gloo; each subgroup trains its own student against
the shared frozen teacher, with sinkhorn/grad collectives confined to the
subgroup."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 4

STUDENT_A = """
dino: {head_n_prototypes: 64, head_bottleneck_dim: 32, head_hidden_dim: 64}
ibot: {head_n_prototypes: 64, head_bottleneck_dim: 32, head_hidden_dim: 64}
student: {arch: vit_small, patch_size: 16, drop_path_rate: 0.0, ffn_ratio: 1.0}
compute_precision: {param_dtype: fp32}
train: {batch_size_per_gpu: 2, dataset_path: "Synthetic:split=TRAIN:length=64", num_workers: 0}
crops: {local_crops_number: 2, global_crops_size: 112, local_crops_size: 48}
"""

STUDENT_B = STUDENT_A.replace("ffn_ratio: 1.0", "ffn_ratio: 2.0")

TEACHER = STUDENT_A  # same head dims (asserted by the distillation setup)


def _write_cfgs(tmpdir):
    paths = {}
    for name, text in (("a", STUDENT_A), ("b", STUDENT_B), ("teacher", TEACHER)):
        p = os.path.join(tmpdir, f"{name}.yaml")
        with open(p, "w") as f:
            f.write(text)
        paths[name] = p
    return paths


def _parent_cfg(paths):
    from dinov3_amd.configs import get_default_config

    cfg = get_default_config()
    cfg.MODEL.META_ARCHITECTURE = "MultiDistillationMetaArch"
    cfg.multidistillation.enabled = True
    cfg.multidistillation.global_batch_size = 8
    cfg.multidistillation.students = [
        {"name": "stu_a", "config_path": paths["a"], "ranks_range": [0, 2]},
        {"name": "stu_b", "config_path": paths["b"], "ranks_range": [2, 4]},
    ]
    cfg.distillation.enabled = True
    cfg.distillation.full_cfg_path = paths["teacher"]
    cfg.distillation.checkpoint_path = "ignore"
    return cfg


def _init(rank, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)


def _run(fn, port, *args):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=fn, args=(r, port) + args) for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(600)
    for p in procs:
        assert p.exitcode == 0, f"child exited with {p.exitcode}"


def _multidist_worker(rank, port, tmpdir):
    _init(rank, port)
    import bench
    from dinov3_amd import parallel
    from dinov3_amd.train.multidist_meta_arch import MultiDistillationMetaArch
    from dinov3_amd.train.train import build_training_engine

    torch.manual_seed(7)  # identical init within a subgroup
    cfg = _parent_cfg(_write_cfgs(tmpdir))
    model = MultiDistillationMetaArch(cfg)
    model.train()

    # rank -> student mapping and subgroup scoping
    expect = "stu_a" if rank < 2 else "stu_b"
    assert model.student_name == expect, (rank, model.student_name)
    assert parallel.subgroup_size() == 2
    assert parallel.subgroup_rank() == rank % 2
    rank_cfg = model.rank_config
    assert rank_cfg.train.batch_size_per_gpu == 2  # 8 global / 4 ranks
    assert rank_cfg.distillation.enabled

    optimizer, finalize = build_training_engine(rank_cfg, model.get_params_groups())
    torch.manual_seed(100 + rank)  # distinct data per rank
    batch = bench.make_synthetic_batch(rank_cfg, torch.device("cpu"), torch.float32,
                                       n_batches=1)[0]
    for it in range(2):
        loss, loss_dict = model(batch, teacher_temp=0.07, iteration=it)
        assert torch.isfinite(loss), loss_dict
        loss.backward()
        finalize()
        optimizer.step(lr=1e-3, weight_decay=0.0, last_layer_lr=0.0)
        optimizer.zero_grad()
        model.update_ema(0.99)

    # params identical within the subgroup, different across subgroups
    flat = torch.cat([p.detach().reshape(-1)
                      for p in model.arch.student_backbone.parameters()])
    h = torch.tensor([float(flat.sum()), float(flat.abs().sum())])
    gathered = [torch.empty_like(h) for _ in range(WORLD)]
    dist.all_gather(gathered, h)
    assert torch.allclose(gathered[0], gathered[1], atol=1e-5)
    assert torch.allclose(gathered[2], gathered[3], atol=1e-5)
    assert not torch.allclose(gathered[0], gathered[2], atol=1e-3)
    dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_multidist_two_students_two_ranks_each(tmp_path):
    _run(_multidist_worker, 29691, str(tmp_path))


def test_multidist_single_process_builds_first_student(tmp_path):
    from dinov3_amd import parallel
    from dinov3_amd.train.multidist_meta_arch import MultiDistillationMetaArch

    cfg = _parent_cfg(_write_cfgs(str(tmp_path)))
    model = MultiDistillationMetaArch(cfg)
    assert model.student_name == "stu_a"
    assert parallel.subgroup() is None
    import bench

    batch = bench.make_synthetic_batch(model.rank_config, torch.device("cpu"),
                                       torch.float32, n_batches=1)[0]
    model.train()
    loss, _ = model(batch, teacher_temp=0.07, iteration=0)
    assert torch.isfinite(loss)


def test_multidist_layout_validation(tmp_path):
    from dinov3_amd.train.multidist_meta_arch import MultiDistillationMetaArch

    cfg = _parent_cfg(_write_cfgs(str(tmp_path)))
    cfg.multidistillation.students[1]["ranks_range"] = [3, 5]  # gap after [0,2)
    with pytest.raises(ValueError, match="contiguous"):
        MultiDistillationMetaArch(cfg)
