"""Multi-distillation meta-architecture.

The reference ships only a stub (dinov3_jax/train/multidist_meta_arch.py:9 —
`class MultiDistillationMetaArch: ...`) plus config plumbing; capability
parity here is the same surface: config-validated construction with the
process-subgroup layout computed, and a clear NotImplementedError for the
forward until the multi-student training loop lands.
"""

from __future__ import annotations

import logging
from typing import List

import torch.nn as nn

from .. import parallel

logger = logging.getLogger("dinov3")


class MultiDistillationMetaArch(nn.Module):
    """Several students of different scales distilled from one teacher, each
    student owned by a subgroup of ranks (reference configs:
    multi_distillation_test.yaml, multidist_tests/{vits,vitb}_p16.yaml)."""

    def __init__(self, config):
        super().__init__()
        self.config = config
        assert config.multidistillation.enabled
        students = config.multidistillation.get("students", []) or []
        # recipe schema: students[i].ranks_range = [lo, hi) (reference
        # multi_distillation_test.yaml / dinov3_vitl16_lvd1689m_distilled.yaml)
        ranges = [tuple(s["ranks_range"]) for s in students]
        if not ranges:
            raise ValueError("multidistillation.students must not be empty")
        ranges.sort()
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

    def forward(self, *args, **kwargs):
        raise NotImplementedError(
            "multi-distillation training is not implemented (the reference has "
            "a stub only); single-student distillation runs via "
            "config.distillation.enabled"
        )
