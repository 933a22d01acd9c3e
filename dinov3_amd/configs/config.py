"""Config system: YAML tree + 3-level merge (defaults <- file <- CLI dotlist).

Schema-compatible replacement for the reference's OmegaConf-based system
(reference: dinov3_jax/configs/config.py:67-146). We implement the same public
surface — get_default_config / setup_config / setup_job /
apply_scaling_rules_to_cfg / write_config — on a small dot-accessible dict so
that the framework has no OmegaConf dependency.
"""

from __future__ import annotations

import logging
import math
import os
from typing import Any, List, Optional

import yaml

logger = logging.getLogger("dinov3")

_DEFAULT_CONFIG_PATH = os.path.join(os.path.dirname(__file__), "ssl_default_config.yaml")


class DotDict(dict):
    """dict with attribute access and struct-strict merge semantics."""

    def __getattr__(self, key: str) -> Any:
        try:
            return self[key]
        except KeyError as e:
            raise AttributeError(key) from e

    def __setattr__(self, key: str, value: Any) -> None:
        self[key] = value

    def __delattr__(self, key: str) -> None:
        del self[key]

    @staticmethod
    def from_nested(obj: Any) -> Any:
        if isinstance(obj, dict):
            return DotDict({k: DotDict.from_nested(v) for k, v in obj.items()})
        if isinstance(obj, list):
            return [DotDict.from_nested(v) for v in obj]
        return obj

    def to_plain(self) -> Any:
        def conv(o):
            if isinstance(o, dict):
                return {k: conv(v) for k, v in o.items()}
            if isinstance(o, list):
                return [conv(v) for v in o]
            return o

        return conv(self)


def _merge_into(dst: DotDict, src: dict, path: str = "", strict: bool = True) -> None:
    """Recursive merge of `src` into `dst`.

    strict=True rejects keys absent from `dst` (struct mode), matching the
    reference's OmegaConf struct-strict behavior, except that a whole new
    sub-tree may be introduced under a `null` leaf.
    """
    for key, value in src.items():
        full = f"{path}.{key}" if path else key
        if key not in dst:
            if strict:
                raise KeyError(f"Unknown config key: {full}")
            logger.warning("config: accepting unknown key %s", full)
        cur = dst.get(key)
        if isinstance(cur, dict) and isinstance(value, dict):
            _merge_into(cur, value, full, strict)
        elif isinstance(value, dict):
            dst[key] = DotDict.from_nested(value)
        else:
            dst[key] = value


def load_yaml(path: str) -> DotDict:
    with open(path) as f:
        data = yaml.safe_load(f) or {}
    return DotDict.from_nested(data)


def get_default_config() -> DotDict:
    return load_yaml(_DEFAULT_CONFIG_PATH)


def _parse_dotlist_value(raw: str) -> Any:
    try:
        return yaml.safe_load(raw)
    except yaml.YAMLError:
        return raw


def apply_dotlist(cfg: DotDict, dotlist: List[str], strict: bool = True) -> None:
    """Apply `a.b.c=value` overrides (CLI)."""
    for item in dotlist:
        if "=" not in item:
            raise ValueError(f"Malformed dotlist entry (expected key=value): {item}")
        key, raw = item.split("=", 1)
        parts = key.strip().split(".")
        node = cfg
        for p in parts[:-1]:
            if p not in node:
                if strict:
                    raise KeyError(f"Unknown config key: {key}")
                node[p] = DotDict()
            node = node[p]
            if not isinstance(node, dict):
                raise KeyError(f"Config key path crosses a leaf: {key}")
        leaf = parts[-1]
        if strict and leaf not in node:
            raise KeyError(f"Unknown config key: {key}")
        node[leaf] = _parse_dotlist_value(raw)


def apply_scaling_rules_to_cfg(cfg: DotDict) -> DotDict:
    """Batch-size lr scaling (reference: configs/config.py:43-56)."""
    if cfg.optim.get("scaling_rule") in (None, "", "none"):
        return cfg
    import torch.distributed as dist

    world_size = dist.get_world_size() if dist.is_initialized() else int(os.environ.get("WORLD_SIZE", 1))
    global_batch = cfg.train.batch_size_per_gpu * world_size
    rule = cfg.optim.scaling_rule
    base_lr = cfg.optim.lr
    if rule == "linear_wrt_256":
        cfg.optim.lr = base_lr * global_batch / 256.0
    elif rule == "sqrt_wrt_1024":
        cfg.optim.lr = base_lr * math.sqrt(global_batch / 1024.0)
    else:
        raise NotImplementedError(f"Unknown scaling rule: {rule}")
    logger.info("scaling rule %s: lr %g -> %g (global batch %d)", rule, base_lr, cfg.optim.lr, global_batch)
    return cfg


def write_config(cfg: DotDict, output_dir: str, name: str = "config.yaml") -> str:
    os.makedirs(output_dir, exist_ok=True)
    path = os.path.join(output_dir, name)
    with open(path, "w") as f:
        yaml.safe_dump(cfg.to_plain(), f, sort_keys=False)
    logger.info("config written to %s", path)
    return path


def setup_config(args, strict_cfg: bool = True, apply_scaling: bool = True) -> DotDict:
    """defaults <- --config-file <- CLI dotlist, then scaling rules + dump."""
    cfg = get_default_config()
    config_file = getattr(args, "config_file", None)
    if config_file:
        # file merges tolerate (and warn on) unknown keys so that configs
        # written for other revisions of the schema still load; CLI dotlists
        # stay strict to catch typos.
        _merge_into(cfg, load_yaml(config_file).to_plain(), strict=False)
    opts = getattr(args, "opts", None) or []
    apply_dotlist(cfg, opts, strict=strict_cfg)
    output_dir = getattr(args, "output_dir", None)
    if output_dir:
        cfg.train.output_dir = output_dir
    if apply_scaling:
        apply_scaling_rules_to_cfg(cfg)
    if cfg.train.output_dir and cfg.train.output_dir not in (".", "./"):
        # skip the config echo for the degenerate cwd output dir (the schema
        # default) so library/test usage doesn't litter the working tree
        try:
            write_config(cfg, cfg.train.output_dir)
        except OSError:
            logger.warning("could not write config to %s", cfg.train.output_dir)
    return cfg


def exit_job(distributed_enabled: bool = True, logging_enabled: bool = True) -> None:
    """Teardown counterpart of setup_job (reference configs parity: the
    run/init.py job_context calls exit_job on the way out)."""
    import logging as _logging

    from .. import parallel

    if distributed_enabled:
        parallel.destroy()
    if logging_enabled:
        for h in _logging.getLogger("dinov3").handlers:
            h.flush()


def setup_job(output_dir: Optional[str] = None, seed: int = 0, distributed_enabled: bool = True,
              logging_enabled: bool = True) -> None:
    """Job context: logging + RNG seeding (+ torch.distributed if launched via torchrun)."""
    from ..logging import setup_logging
    from ..utils.utils import fix_random_seeds
    from .. import parallel

    if distributed_enabled:
        parallel.enable_distributed()
    if logging_enabled:
        setup_logging(output=output_dir)
    fix_random_seeds(seed + parallel.get_rank())
