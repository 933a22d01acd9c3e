"""World-size-independent optimizer-state interchange.

Both engines key their state by the fused param-group's ordered name tuple
(param order inside a group is identical — both come from
``fuse_params_groups``). The canonical form flattens each group's fp32
exp_avg / exp_avg_sq / master into ONE unpadded vector in param-concatenation
order, so any saved layout can be resliced into any live layout:

* FusedAdamW ("groups" format: per-param tensor lists, replicated) — one
  rank's state suffices.
* ShardedEngine ("shards" format: this rank's slice of the padded flat
  bucket) — needs every saved rank's state; slices are concatenated in rank
  order and the world-size-dependent padding is trimmed via the recorded
  unpadded "total".

The reference's orbax restore is shape-driven partial restore
(dinov3_jax/checkpointer/checkpointer.py:157-184); this is the sharded-torch
equivalent that lets an 8-GPU run resume a 1-GPU checkpoint and vice versa.
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Tuple

import torch

logger = logging.getLogger("dinov3")

Canonical = Dict[str, object]  # {"step_count": int, "groups": {names: fields}}


def state_format(state: dict) -> str:
    return "shards" if "shards" in state else "groups"


def _flat(tensors) -> torch.Tensor:
    return torch.cat([t.reshape(-1).float() for t in tensors])


def to_canonical(states: List[dict]) -> Canonical:
    """Build the canonical form from saved optimizer state dict(s).

    `states`: one state for the replicated "groups" format, or the states of
    ALL saved ranks (any order) for the "shards" format."""
    s0 = states[0]
    out: Canonical = {"step_count": s0["step_count"], "groups": {}}
    if state_format(s0) == "groups":
        for gs in s0["groups"]:
            out["groups"][tuple(gs["names"])] = {
                "exp_avg": _flat(gs["exp_avg"]),
                "exp_avg_sq": _flat(gs["exp_avg_sq"]),
                "master": _flat(gs["master"]) if gs["master"] is not None else None,
            }
        return out

    world = s0["world"]
    by_rank = {s["rank"]: s for s in states}
    assert len(by_rank) == world, (
        f"resharding a world-{world} checkpoint needs all {world} ranks' "
        f"optimizer states, got ranks {sorted(by_rank)}")
    ordered = [by_rank[r] for r in range(world)]
    for i, shard0 in enumerate(s0["shards"]):
        key = tuple(shard0["names"])
        total = shard0.get("total")

        def cat(field):
            parts = [s["shards"][i][field] for s in ordered]
            if any(p is None for p in parts):
                return None
            v = torch.cat([p.reshape(-1).float().cpu() for p in parts])
            return v[:total] if total is not None else v

        out["groups"][key] = {"exp_avg": cat("exp_avg"),
                              "exp_avg_sq": cat("exp_avg_sq"),
                              "master": cat("master")}
    return out


def load_optimizer_state(optimizer, ckpt_dir, payload: dict, rank: int, world: int) -> None:
    """Restore `payload["optimizer"]` into `optimizer`, resharding when the
    saved layout (engine type or world size) differs from the live one.
    `ckpt_dir` is the checkpoint directory (or a single rank file)."""
    from pathlib import Path

    opt_state = payload.get("optimizer")
    if opt_state is None:
        return
    fmt = state_format(opt_state)
    want = getattr(optimizer, "state_format", "groups")
    saved_world = opt_state.get("world", payload.get("world_size", 1))
    if fmt == want and (fmt == "groups" or saved_world == world):
        optimizer.load_state_dict(opt_state)
        return
    if fmt == "shards":
        base = Path(ckpt_dir)
        if base.is_file():
            base = base.parent
        states = []
        for r in range(saved_world):
            if r == opt_state.get("rank"):
                states.append(opt_state)
                continue
            path = base / f"rank_{r}.pth"
            assert path.exists(), (
                f"resharded restore needs {path} (world-{saved_world} checkpoint)")
            states.append(torch.load(path, map_location="cpu", weights_only=False)["optimizer"])
    else:
        states = [opt_state]
    logger.info("resharding optimizer state: saved %s/world %d -> live %s/world %d",
                fmt, saved_world, want, world)
    optimizer.load_canonical(to_canonical(states))
