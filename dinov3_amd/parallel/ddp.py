"""Bucketed data-parallel gradient reduction over RCCL (xGMI).

Grad hooks pack gradients into flat buckets as backward produces them and
launch an async all-reduce per full bucket on a dedicated comm stream, so
communication overlaps the remaining backward (SURVEY §2.3 C2/C3 mapping for
the non-sharded path). Bucket size defaults to 32 MiB — sized for 7x153 GB/s
point-to-point xGMI links where ring all-reduce is per-link bound: fewer,
larger transfers amortize latency without delaying the first reduction.

Reduction math: sum then divide by world (pmean), in the grad dtype of the
params (reduce_dtype fp32 upcast is applied for bf16 grads).
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

logger = logging.getLogger("dinov3")


class GradReducer:
    def __init__(self, params: List[torch.nn.Parameter], bucket_cap_mb: float = 32.0,
                 reduce_dtype: Optional[torch.dtype] = None):
        from . import subgroup, subgroup_size

        self.params = [p for p in params if p.requires_grad]
        self.group = subgroup()
        self.world = subgroup_size()
        self.reduce_dtype = reduce_dtype
        self._works: List[dist.Work] = []
        self._hooks = []
        self._bucket_cap = int(bucket_cap_mb * 1024 * 1024)
        self._pending: List[torch.Tensor] = []
        self._pending_bytes = 0
        self._use_hooks = self.world > 1
        if self._use_hooks:
            for p in self.params:
                h = p.register_post_accumulate_grad_hook(self._on_grad)
                self._hooks.append(h)

    def _on_grad(self, p: torch.nn.Parameter) -> None:
        if p.grad is None:
            return
        self._pending.append(p.grad)
        self._pending_bytes += p.grad.numel() * p.grad.element_size()
        if self._pending_bytes >= self._bucket_cap:
            self._flush()

    def _flush(self) -> None:
        if not self._pending:
            return
        grads = self._pending
        self._pending = []
        self._pending_bytes = 0
        flat = torch.cat([g.reshape(-1) for g in grads])
        if self.reduce_dtype is not None and flat.dtype != self.reduce_dtype:
            flat = flat.to(self.reduce_dtype)
        work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.group, async_op=True)
        self._works.append((work, flat, grads))

    def finalize(self) -> None:
        """Wait for all reductions and scatter the averaged grads back."""
        if not self._use_hooks:
            return
        self._flush()
        for work, flat, grads in self._works:
            work.wait()
            flat = flat.div_(self.world)
            offset = 0
            for g in grads:
                n = g.numel()
                g.copy_(flat[offset: offset + n].reshape(g.shape).to(g.dtype))
                offset += n
        self._works = []

    def remove_hooks(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks = []


def all_reduce_scalar_sums(values: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """All-reduce a dict of scalar tensors (sum). Used for grad-norm sq."""
    from . import subgroup, subgroup_size

    if not (dist.is_available() and dist.is_initialized()) or subgroup_size() == 1:
        return values
    keys = sorted(values.keys())
    device = next(iter(values.values())).device if values else torch.device("cpu")
    buf = torch.stack([values[k].to(device).float().reshape(()) for k in keys])
    dist.all_reduce(buf, group=subgroup())
    return {k: buf[i] for i, k in enumerate(keys)}
