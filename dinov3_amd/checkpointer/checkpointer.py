"""Checkpointing: same directory layout + iteration semantics as the
reference (output_dir/ckpt/<iteration>/, keep-last-N + keep_every hardlink
copies, partial restore) on torch-native per-rank shard files.

Parity: dinov3_jax/checkpointer/checkpointer.py:23-192.
"""

from __future__ import annotations

import logging
import os
import re
import shutil
from enum import Enum
from pathlib import Path
from typing import Any, Dict, Optional

import torch

from .. import parallel

logger = logging.getLogger("dinov3")


class CheckpointRetentionPolicy(Enum):
    ALL = "all"
    LAST = "last"
    NONE = "none"


def _ckpt_root(output_dir: str) -> Path:
    return Path(output_dir) / "ckpt"


def find_latest_checkpoint(output_dir: str) -> Optional[Path]:
    root = _ckpt_root(output_dir)
    if not root.is_dir():
        return None
    best: Optional[Path] = None
    best_iter = -1
    for child in root.iterdir():
        if child.is_dir() and re.fullmatch(r"\d+", child.name):
            it = int(child.name)
            if it > best_iter and (child / ".complete").exists():
                best_iter = it
                best = child
    return best


def keep_last_n_checkpoints(output_dir: str, n: int, keep_every: int = 0) -> None:
    root = _ckpt_root(output_dir)
    if not root.is_dir() or not parallel.is_main_process():
        return
    iters = sorted(
        int(c.name) for c in root.iterdir() if c.is_dir() and re.fullmatch(r"\d+", c.name)
    )
    for it in iters[:-n] if n > 0 else []:
        if keep_every and it % keep_every == 0:
            continue
        shutil.rmtree(root / str(it), ignore_errors=True)
        logger.info("removed old checkpoint %d", it)


def save_checkpoint(output_dir: str, iteration: int, model: torch.nn.Module,
                    optimizer: Optional[Any] = None, extra: Optional[Dict[str, Any]] = None,
                    max_to_keep: int = 3, keep_every: int = 0,
                    skip_prefixes: Optional[Any] = None) -> Path:
    """Each rank writes its own shard file (rank0 also writes metadata).

    skip_prefixes: state-dict key prefixes to omit — the
    `register_dont_save_hooks` analogue the reference README promises
    (dinov3_jax/README.md:35, call site train.py:453-457 is dead); used to
    skip the frozen distillation teacher, which never changes."""
    ckpt_dir = _ckpt_root(output_dir) / str(iteration)
    ckpt_dir.mkdir(parents=True, exist_ok=True)
    rank = parallel.get_rank()
    state = model.state_dict()
    if skip_prefixes:
        state = {k: v for k, v in state.items()
                 if not any(k.startswith(p) for p in skip_prefixes)}
    payload: Dict[str, Any] = {
        "iteration": iteration,
        "model": state,
        "world_size": parallel.get_world_size(),
    }
    if optimizer is not None:
        payload["optimizer"] = optimizer.state_dict()
    if extra:
        payload["extra"] = extra
    torch.save(payload, ckpt_dir / f"rank_{rank}.pth")
    parallel.barrier()
    if parallel.is_main_process():
        (ckpt_dir / ".complete").touch()
        keep_last_n_checkpoints(output_dir, max_to_keep, keep_every)
    parallel.barrier()
    logger.info("saved checkpoint %s", ckpt_dir)
    return ckpt_dir


def load_checkpoint(ckpt_dir: os.PathLike, model: torch.nn.Module,
                    optimizer: Optional[Any] = None, strict: bool = True) -> Dict[str, Any]:
    ckpt_dir = Path(ckpt_dir)
    if ckpt_dir.is_file():  # MODEL.WEIGHTS may point at a rank file directly
        path = ckpt_dir
    else:
        rank = parallel.get_rank()
        path = ckpt_dir / f"rank_{rank}.pth"
        if not path.exists():
            path = ckpt_dir / "rank_0.pth"
    payload = torch.load(path, map_location="cpu", weights_only=False)
    missing, unexpected = model.load_state_dict(payload["model"], strict=strict)
    if missing or unexpected:
        logger.warning("partial restore: missing=%d unexpected=%d", len(missing), len(unexpected))
    if optimizer is not None and "optimizer" in payload:
        # Resharded restore: the saved engine layout / world size may differ
        # from the live one (e.g. an 8-GPU run resuming a 1-GPU checkpoint)
        from ..train.optim_state import load_optimizer_state

        load_optimizer_state(optimizer, ckpt_dir, payload,
                             rank=parallel.get_rank(), world=parallel.get_world_size())
    logger.info("loaded checkpoint %s (iteration %d)", ckpt_dir, payload["iteration"])
    return payload
