"""Cached device-side plans for the single-launch multi-tensor kernels.

A plan freezes the (pointer table, sizes, chunk table) of a tensor list into
device memory once; each step then costs one kernel launch with zero
host-side table building. Pointer stability is re-checked cheaply each call
and the plan rebuilt if any tensor re-allocated.
"""

from __future__ import annotations

from typing import Sequence

import torch

CHUNK = 65536


class MultiTensorPlan:
    def __init__(self, tensor_lists: Sequence[Sequence[torch.Tensor]]):
        self.lists = [list(l) for l in tensor_lists]
        self.n_tensors = len(self.lists[0])
        self.device = self.lists[0][0].device
        self.is_bf16 = self.lists[0][0].dtype == torch.bfloat16
        sizes = [t.numel() for t in self.lists[0]]
        ct, co = [], []
        for i, s in enumerate(sizes):
            for off in range(0, s, CHUNK):
                ct.append(i)
                co.append(off)
        self.sizes = torch.tensor(sizes, dtype=torch.int64, device=self.device)
        self.ct = torch.tensor(ct, dtype=torch.int32, device=self.device)
        self.co = torch.tensor(co, dtype=torch.int64, device=self.device)
        self._ptr_list = [t.data_ptr() for lst in self.lists for t in lst]
        self.ptrs = torch.tensor(self._ptr_list, dtype=torch.int64, device=self.device)

    def check_pointers(self) -> bool:
        cur = [t.data_ptr() for lst in self.lists for t in lst]
        if cur == self._ptr_list:
            return True
        self._ptr_list = cur
        self.ptrs = torch.tensor(cur, dtype=torch.int64, device=self.device)
        return False


def ema_planned(plan: MultiTensorPlan, momentum: float) -> None:
    from . import hip_ops

    plan.check_pointers()
    hip_ops().multi_tensor_ema_planned(plan.ptrs, plan.sizes, plan.ct, plan.co,
                                       plan.n_tensors, momentum, plan.is_bf16)


def adamw_planned(plan: MultiTensorPlan, lr_mult: torch.Tensor, wd_mult: torch.Tensor,
                  is_last: torch.Tensor, sub_id: torch.Tensor, clip: torch.Tensor,
                  lr: float, last_lr: float, wd: float, beta1: float, beta2: float,
                  eps: float, step: int, has_master: bool) -> None:
    from . import hip_ops

    plan.check_pointers()
    bc1 = 1.0 - beta1**step
    bc2 = 1.0 - beta2**step
    grad_is_f32 = plan.lists[1][0].dtype == torch.float32
    hip_ops().multi_tensor_adamw_planned(
        plan.ptrs, plan.sizes, plan.ct, plan.co, plan.n_tensors, lr_mult, wd_mult,
        is_last, sub_id, clip, lr, last_lr, wd, beta1, beta2, eps, bc1, bc2, has_master,
        plan.is_bf16, grad_is_f32,
    )


def l2norm_planned(plan: MultiTensorPlan, sub_id: torch.Tensor, n_submodels: int,
                   list_index: int = 0) -> torch.Tensor:
    """Per-submodel sum of squared values over plan list `list_index`."""
    from . import hip_ops

    plan.check_pointers()
    n = plan.n_tensors
    ptrs = plan.ptrs[list_index * n: (list_index + 1) * n]
    is_bf16 = plan.lists[list_index][0].dtype == torch.bfloat16
    return hip_ops().multi_tensor_l2norm_planned(ptrs, plan.sizes, plan.ct, plan.co,
                                                 sub_id, n_submodels, is_bf16)
