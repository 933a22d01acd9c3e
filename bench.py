#!/usr/bin/env python
"""Benchmark driver contract: flagship DINOv3 pretraining step.

`python bench.py --gpus N --steps K --warmup W` runs the BASELINE.json
headline config — ViT-L/16 DINOv3 pretrain @224px, batch 64/GPU, bf16,
synthetic data, random-init weights — and prints ONE JSON line from rank 0
with the whole-job images/sec.

For N>1 the driver launches via torch.distributed.run (one rank per GPU over
RCCL); RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* come from the env.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time
import torch

REPO_ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO_ROOT)

# Pre-tuned hipBLASLt algorithm selections for the ViT-L step's GEMM shapes
# (PyTorch TunableOp, tuned on MI355X — +4% step time). Tuning itself stays
# off; delete the env vars to disable.
_TUNED_DIR = os.path.join(REPO_ROOT, "dinov3_amd", "tunableop")
if os.path.isdir(_TUNED_DIR):
    # torch appends the device ordinal before .csv: ship tunableop{0..7}.csv
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                          os.path.join(_TUNED_DIR, "tunableop.csv"))

BASELINE_IMG_PER_SEC_PER_GPU = 2048 / 0.57 / 32  # Meta RSC anchor: 0.57 s/iter @ 2048 global batch, 32 GPUs


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch-size", type=int, default=64, help="images per GPU")
    p.add_argument("--arch", type=str, default="vit_large",
                   help="override arch (non-default invalidates the headline number)")
    p.add_argument("--local-crops", type=int, default=8)
    p.add_argument("--global-size", type=int, default=224)
    p.add_argument("--local-size", type=int, default=96)
    p.add_argument("--patch-size", type=int, default=16)
    p.add_argument("--grad-checkpointing", action="store_true",
                   help="activation checkpointing (config #5: ViT-g sizing)")
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--data", type=str, default="synthetic",
                   choices=["synthetic", "loader", "gpu-aug"],
                   help="synthetic = pre-staged device batches (headline); loader = "
                        "real DataLoader path with per-step augment+collate+H2D in the "
                        "timed region; gpu-aug = workers decode only, multi-crop "
                        "augmentation batched on the training GPU")
    p.add_argument("--num-workers", type=int, default=12,
                   help="dataloader workers per rank for --data loader")
    p.add_argument("--profile-tag", type=str, default="", help="label kernels for rocprof runs")
    return p.parse_args()


def build_cfg(args):
    from dinov3_amd.configs import get_default_config, DotDict

    cfg = get_default_config()
    cfg.student.arch = args.arch
    cfg.student.patch_size = args.patch_size
    if args.grad_checkpointing:
        cfg.train.checkpointing = True
    cfg.student.drop_path_rate = 0.3
    cfg.student.layerscale = 1.0e-05
    cfg.train.batch_size_per_gpu = args.batch_size
    cfg.train.centering = "sinkhorn_knopp"
    cfg.crops.local_crops_number = args.local_crops
    cfg.crops.global_crops_size = args.global_size
    cfg.crops.local_crops_size = args.local_size
    cfg.compute_precision.param_dtype = args.dtype
    cfg.optim.clip_grad = 3.0
    return cfg


def make_synthetic_batch(cfg, device, dtype, n_batches=4):
    """Pregenerated synthetic collated batches (shape-exact, random data)."""
    from dinov3_amd.data import MaskingGenerator, collate_data_and_cast
    from dinov3_amd.data.datasets import random_image

    B = cfg.train.batch_size_per_gpu
    gs, ls = cfg.crops.global_crops_size, cfg.crops.local_crops_size
    n_local = cfg.crops.local_crops_number
    p = cfg.student.patch_size
    n_tokens = (gs // p) ** 2
    mask_gen = MaskingGenerator(input_size=(gs // p, gs // p), max_num_patches=int(0.5 * n_tokens))
    batches = []
    for _ in range(n_batches):
        samples = []
        for _ in range(B):
            sample = {
                "global_crops": [torch.randn(3, gs, gs) for _ in range(2)],
                "local_crops": [torch.randn(3, ls, ls) for _ in range(n_local)],
            }
            samples.append((sample, ()))
        batch = collate_data_and_cast(
            samples, mask_ratio_tuple=tuple(cfg.ibot.mask_ratio_min_max),
            mask_probability=cfg.ibot.mask_sample_probability, dtype=dtype,
            n_tokens=n_tokens, mask_generator=mask_gen,
        )
        batches.append({k: (v.to(device) if isinstance(v, torch.Tensor) else v) for k, v in batch.items()})
    return batches


def main():
    args = parse_args()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")

    from dinov3_amd import parallel
    from dinov3_amd.train.ssl_meta_arch import SSLMetaArch
    from dinov3_amd.train.train import DTYPE_MAP, build_training_engine

    parallel.enable_distributed()
    rank = parallel.get_rank()
    world = parallel.get_world_size()
    device = parallel.device()
    use_gpu = device.type == "cuda"
    if use_gpu:
        torch.cuda.set_device(device)

    torch.manual_seed(1234 + rank)
    cfg = build_cfg(args)
    dtype = DTYPE_MAP[args.dtype] if use_gpu else torch.float32

    model = SSLMetaArch(cfg)
    model = model.to(device=device, dtype=dtype)
    model.train()

    groups = model.get_params_groups()
    optimizer, finalize_backward = build_training_engine(cfg, groups)

    if args.data == "gpu-aug":
        from dinov3_amd.data.gpu_pipeline import build_gpu_augment_pipeline_from_cfg

        cfg.train.num_workers = args.num_workers
        cfg.train.dataset_path = "Synthetic:split=TRAIN"
        pipeline = build_gpu_augment_pipeline_from_cfg(cfg, device, dtype)
        pipe_it = iter(pipeline)

        def next_batch(i):
            nonlocal pipe_it
            try:
                return next(pipe_it)
            except StopIteration:
                pipe_it = iter(pipeline)
                return next(pipe_it)
    elif args.data == "loader":
        # full input path in the timed region: synthetic decode -> multi-crop
        # augment (workers) -> collate -> pinned H2D each step
        from dinov3_amd.train.train import batch_to_device, build_data_loader_from_cfg

        cfg.train.num_workers = args.num_workers
        cfg.train.dataset_path = "Synthetic:split=TRAIN"
        loader = build_data_loader_from_cfg(cfg, model)

        def cycle():
            while True:
                yield from loader

        if use_gpu:
            from dinov3_amd.data.prefetch import CudaBatchPrefetcher

            loader_it = CudaBatchPrefetcher(cycle(), device)

            def next_batch(i):
                return next(loader_it)
        else:
            loader_it = cycle()

            def next_batch(i):
                return batch_to_device(next(loader_it), device)
    else:
        batches = make_synthetic_batch(cfg, device, dtype)

        def next_batch(i):
            return batches[i % len(batches)]
    clip = cfg.optim.clip_grad

    def step(i):
        data = next_batch(i)
        loss, _ = model(data, teacher_temp=0.07, iteration=i)
        loss.backward()
        finalize_backward()
        clip_scales = None
        if clip:
            sums = optimizer.grad_norm_sums()
            # only shard-local sums need the reduction (DDP grads are already
            # replicated — reducing again would inflate norms by world_size)
            if world > 1 and getattr(optimizer, "needs_norm_allreduce", True):
                import torch.distributed as dist

                dist.all_reduce(sums)
            clip_scales = optimizer.clip_factors(sums, clip)
        optimizer.step(lr=1e-4, weight_decay=0.04, last_layer_lr=0.0, clip_scales=clip_scales)
        optimizer.zero_grad()
        model.update_ema(0.992)
        return loss

    for i in range(args.warmup):
        step(i)

    parallel.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(args.warmup + i)
    parallel.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64, device=device if use_gpu else "cpu")
    if world > 1:
        import torch.distributed as dist

        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    images = args.batch_size * world * args.steps
    img_per_sec = images / elapsed
    ms_per_step = elapsed / args.steps * 1000.0
    if use_gpu:
        peak_gb = torch.cuda.max_memory_allocated(device) / 2**30
        print(f"[bench] peak memory allocated: {peak_gb:.1f} GiB "
              f"(rank {rank}, 288 GiB HBM3E)", file=sys.stderr)

    if rank == 0:
        parallelism = ("fsdp" if (world > 1 and cfg.compute_precision.sharding_strategy
                                  in ("SHARD_GRAD_OP", "FULL_SHARD")) else "dp")
        result = {
            "metric": "images/sec (whole node) ViT-L/16 DINOv3 pretrain 224px",
            "value": img_per_sec,
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": img_per_sec / (BASELINE_IMG_PER_SEC_PER_GPU * world),
            "dtype": args.dtype if use_gpu else "fp32",
            "data": {"synthetic": "synthetic",
                     "loader": "synthetic(loader+H2D timed)",
                     "gpu-aug": "synthetic(gpu-augment timed)"}[args.data],
            "config": {
                "model": args.arch,
                "global_batch": args.batch_size * world,
                "img_size": args.global_size,
                "local_crops": args.local_crops,
                "parallelism": f"{parallelism}{world}",
            },
        }
        print(json.dumps(result))
    parallel.destroy()


if __name__ == "__main__":
    main()
