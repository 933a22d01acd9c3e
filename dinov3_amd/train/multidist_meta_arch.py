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
        students = config.multidistillation.get("students", [])
        world = parallel.get_world_size()
        self.subgroup_sizes: List[int] = [s.get("ranks", 1) for s in students] or [world]
        if sum(self.subgroup_sizes) != world:
            raise ValueError(
                f"multidistillation subgroups {self.subgroup_sizes} must cover world size {world}"
            )
        logger.info("multidistillation subgroups: %s", self.subgroup_sizes)

    def forward(self, *args, **kwargs):
        raise NotImplementedError(
            "multi-distillation training is not implemented (the reference has "
            "a stub only); single-student distillation runs via "
            "config.distillation.enabled"
        )
