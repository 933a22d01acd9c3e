"""Schedules as precomputed fp64 arrays indexed by iteration.

Parity: dinov3_jax/train/cosine_lr_scheduler.py (the trunc_extra branch there
references an undefined variable, SURVEY §8 B5 — implemented here as
documented: compute the cosine over (1+trunc_extra)*steps and truncate,
renormalized so the truncated end hits final_value).
"""

from __future__ import annotations

import numpy as np


class CosineScheduler:
    def __init__(self, base_value, final_value, total_iters, warmup_iters=0,
                 start_warmup_value=0, freeze_iters=0, trunc_extra=0.0):
        self.final_value = np.float64(final_value)
        self.total_iters = total_iters
        freeze_iters = min(freeze_iters, total_iters)
        warmup_iters = min(warmup_iters, total_iters - freeze_iters)
        freeze_schedule = np.zeros((freeze_iters,))
        warmup_schedule = np.linspace(start_warmup_value, base_value, warmup_iters)
        cosine_steps = total_iters - warmup_iters - freeze_iters
        if trunc_extra == 0:
            iters = np.arange(cosine_steps)
            schedule = final_value + 0.5 * (base_value - final_value) * (
                1 + np.cos(np.pi * iters / max(len(iters), 1))
            )
        else:
            full = int(round((1 + trunc_extra) * cosine_steps))
            angles = np.linspace(0, np.pi, max(full, 1))[:cosine_steps]
            s = (np.cos(angles) + 1) / 2  # 1 -> s_end
            s = (s - s[-1]) / max(1 - s[-1], 1e-12)  # 1 -> 0 over the truncated range
            schedule = s * (base_value - final_value) + final_value
        self.schedule = np.concatenate([freeze_schedule, warmup_schedule, schedule]).astype(np.float64)
        assert len(self.schedule) == self.total_iters

    def gen(self):
        return self.schedule

    def __getitem__(self, it):
        if it >= self.total_iters:
            return float(self.final_value)
        return float(self.schedule[it])


class linear_warmup_cosine_decay:
    def __init__(self, start, peak, end, warmup_iterations, total_iterations, cosine_iterations=None):
        self.end = np.float64(end)
        linear = np.linspace(start, peak, warmup_iterations, endpoint=False)
        if cosine_iterations is None:
            cosine_iterations = total_iterations - warmup_iterations
        cosine = np.cos(np.linspace(0, np.pi, cosine_iterations))
        cosine = (cosine + 1) / 2
        cosine = (peak - end) * cosine + end
        remaining = total_iterations - cosine_iterations - warmup_iterations
        assert remaining >= 0
        constant = np.full((remaining,), fill_value=end)
        self.schedule = np.concatenate([linear, cosine, constant]).astype(np.float64)

    def gen(self):
        return self.schedule

    def __getitem__(self, it):
        # Past the end of the schedule the value is the terminal `end`, even
        # when a degenerate cosine span (cosine_iterations==1 with
        # warmup≈total) leaves the last stored sample at `peak`.
        if it >= len(self.schedule):
            return float(self.end)
        return float(self.schedule[it])
