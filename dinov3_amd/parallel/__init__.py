"""Distributed runtime: one process per GPU over RCCL (torch.distributed "nccl"
backend on ROCm) on a single node's xGMI fabric, gloo on CPU.

Replaces the reference's JAX mesh/shard_map model (dinov3_jax/distributed/__init__.py:16-21,
train/train.py:322-325) with the PyTorch process model: torchrun sets
RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*, we create one process group and pin each
rank to its GPU.
"""

from __future__ import annotations

import logging
import os
from typing import Optional

import torch
import torch.distributed as dist

logger = logging.getLogger("dinov3")

_INITIALIZED_HERE = False


def is_enabled() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_enabled() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_enabled() else 1


def get_local_rank() -> int:
    if not is_enabled():
        return 0
    return int(os.environ.get("LOCAL_RANK", get_rank() % max(1, torch.cuda.device_count() or 1)))


def is_main_process() -> bool:
    return get_rank() == 0


def enable_distributed(backend: Optional[str] = None, timeout_s: int = 1800) -> None:
    """Initialise torch.distributed if launched under torchrun; no-op otherwise.

    backend defaults to "nccl" (= RCCL over xGMI) when a GPU is visible,
    "gloo" for CPU-only runs (tests).
    """
    global _INITIALIZED_HERE
    if is_enabled():
        return
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return  # single-process run
    import datetime

    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
        torch.cuda.set_device(local_rank)
    dist.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=timeout_s))
    _INITIALIZED_HERE = True
    logger.info("distributed: rank %d / %d (backend %s)", get_rank(), get_world_size(), backend)


def barrier() -> None:
    if is_enabled():
        dist.barrier()


# ---------------------------------------------------------------------------
# Training-collective subgroup (multi-distillation): when set, every
# loss/gradient collective (sinkhorn psums, grad reduce-scatter/all-reduce,
# koleo gather, grad-norm sums) runs over THIS group instead of the world.
# Data sharding (sampler rank striding) and logging stay world-scoped.
_SUBGROUP = None


def set_subgroup(pg) -> None:
    global _SUBGROUP
    _SUBGROUP = pg


def subgroup():
    """Process group for training collectives (None = the world group)."""
    return _SUBGROUP


def subgroup_size() -> int:
    if not is_enabled():
        return 1
    return dist.get_world_size(_SUBGROUP) if _SUBGROUP is not None else dist.get_world_size()


def subgroup_rank() -> int:
    if not is_enabled():
        return 0
    return dist.get_rank(_SUBGROUP) if _SUBGROUP is not None else dist.get_rank()


def destroy() -> None:
    global _INITIALIZED_HERE
    if _INITIALIZED_HERE and is_enabled():
        dist.destroy_process_group()
        _INITIALIZED_HERE = False


def device() -> torch.device:
    if torch.cuda.is_available():
        return torch.device("cuda", get_local_rank())
    return torch.device("cpu")
