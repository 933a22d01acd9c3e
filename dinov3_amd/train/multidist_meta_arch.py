"""Multi-distillation meta-architecture: K students of different scales
trained simultaneously against one frozen teacher, each student owned by a
contiguous subgroup of ranks.

The reference ships only a stub (dinov3_jax/train/multidist_meta_arch.py:9)
plus config plumbing; the intended semantics survive in the copied original
at dinov3_jax/models/temp.py:109-170 (process subgroups per student). Here it
is implemented for real, the MI355X way:

* rank layout from multidistillation.students[i].ranks_range (contiguous,
  covering the world);
* ONE torch.distributed subgroup per student (created collectively on every
  rank), installed as the session's training-collective group
  (parallel.set_subgroup) so every sinkhorn psum, gradient reduce-scatter /
  all-reduce, KoLeo gather and grad-norm sum is scoped to the ranks that
  share the student — data sharding and logging stay world-scoped;
* each rank constructs ONLY its own student as a standard single-student
  distillation SSLMetaArch (frozen teacher from the shared `distillation`
  section) and this wrapper delegates forward/EMA/param-groups/state to it.

Checkpoints keep the uniform rank-file layout: each rank saves its own
student's state (teacher weights are skipped — frozen), so every student
resumes from the same ckpt/<iter>/ directory.
"""

from __future__ import annotations

import logging
from typing import List, Optional

import torch.nn as nn

from .. import parallel

logger = logging.getLogger("dinov3")


def _student_rank_config(config, student_entry):
    """defaults <- student yaml <- shared sections from the parent config."""
    from ..configs import get_default_config, load_yaml
    from ..configs.config import _merge_into

    cfg = get_default_config()
    _merge_into(cfg, load_yaml(student_entry["config_path"]).to_plain(), strict=False)
    # shared orchestration sections come from the parent recipe
    _merge_into(cfg, {"MODEL": config.MODEL.to_plain()
                      if hasattr(config.MODEL, "to_plain") else dict(config.MODEL)},
                strict=False)
    for section in ("distillation", "multidistillation"):
        src = getattr(config, section)
        _merge_into(cfg, {section: src.to_plain() if hasattr(src, "to_plain") else dict(src)},
                    strict=False)
    cfg.MODEL.META_ARCHITECTURE = "SSLMetaArch"  # the inner arch is plain SSL
    cfg.distillation.enabled = True
    return cfg


class MultiDistillationMetaArch(nn.Module):
    """Several students of different scales distilled from one teacher, each
    student owned by a subgroup of ranks (reference configs:
    multi_distillation_test.yaml, multidist_tests/{vits,vitb}_p16.yaml)."""

    def __init__(self, config):
        super().__init__()
        self.config = config
        assert config.multidistillation.enabled
        students = config.multidistillation.get("students", []) or []
        ranges = [tuple(s["ranks_range"]) for s in students]
        if not ranges:
            raise ValueError("multidistillation.students must not be empty")
        order = sorted(range(len(ranges)), key=lambda i: ranges[i])
        ranges = [ranges[i] for i in order]
        students = [students[i] for i in order]
        if ranges[0][0] != 0:
            raise ValueError(f"subgroup ranges must start at rank 0, got {ranges}")
        for (a0, a1), (b0, b1) in zip(ranges, ranges[1:]):
            if a1 != b0:
                raise ValueError(f"subgroup ranges must be contiguous, got {ranges}")
        self.total_ranks = ranges[-1][1]
        self.subgroup_sizes: List[int] = [hi - lo for lo, hi in ranges]
        self.rank_ranges = ranges
        world = parallel.get_world_size()
        if world > 1 and self.total_ranks != world:
            raise ValueError(
                f"multidistillation layout covers {self.total_ranks} ranks "
                f"but world size is {world}")
        logger.info("multidistillation subgroups: %s (total %d ranks)",
                    self.subgroup_sizes, self.total_ranks)

        # ---- this rank's student + collective subgroup ----
        rank = parallel.get_rank()
        if world > 1:
            import torch.distributed as dist

            my_group = None
            my_idx = 0
            for i, (lo, hi) in enumerate(ranges):
                pg = dist.new_group(ranks=list(range(lo, hi)))  # collective
                if lo <= rank < hi:
                    my_group, my_idx = pg, i
            parallel.set_subgroup(my_group)
        else:
            my_idx = 0  # single-process runs build the first student
        self.student_index = my_idx
        self.student_name = students[my_idx].get("name", f"student{my_idx}")

        cfg = _student_rank_config(config, students[my_idx])
        gb = config.multidistillation.get("global_batch_size")
        if gb:
            assert gb % self.total_ranks == 0, (
                f"global_batch_size {gb} must divide over {self.total_ranks} ranks")
            cfg.train.batch_size_per_gpu = gb // self.total_ranks
        self.rank_config = cfg

        from .ssl_meta_arch import SSLMetaArch

        logger.info("multidistillation rank %d -> student '%s' (%s), subgroup of %d",
                    rank, self.student_name, cfg.student.arch,
                    self.subgroup_sizes[my_idx])
        self.arch = SSLMetaArch(cfg)

    # ------------------------------------------------------------- delegate
    @property
    def is_distillation_enabled(self) -> bool:
        return True

    @property
    def gram_use_loss(self):
        return self.arch.gram_use_loss

    @property
    def has_gram_teacher(self):
        return self.arch.has_gram_teacher

    @property
    def gram_it_load_ema_teacher(self):
        return self.arch.gram_it_load_ema_teacher

    def forward(self, data, teacher_temp: float, iteration: int = 0):
        return self.arch(data, teacher_temp=teacher_temp, iteration=iteration)

    def update_ema(self, momentum: float) -> None:
        self.arch.update_ema(momentum)

    def update_gram_teacher(self) -> None:
        self.arch.update_gram_teacher()

    def get_params_groups(self):
        return self.arch.get_params_groups()

    # keep the uniform checkpoint layout: rank files hold the inner arch's
    # keys directly (no "arch." prefix)
    def state_dict(self, *args, **kwargs):
        return self.arch.state_dict(*args, **kwargs)

    def load_state_dict(self, state_dict, strict: bool = True):
        return self.arch.load_state_dict(state_dict, strict=strict)

    def train(self, mode: bool = True):
        self.arch.train(mode)
        return self

    def to(self, *args, **kwargs):
        self.arch = self.arch.to(*args, **kwargs)
        return self
