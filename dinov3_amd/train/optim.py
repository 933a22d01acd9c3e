"""FusedAdamW: multi-tensor AdamW over fused param groups.

Replaces the reference's optax.multi_transform tower (train.py:75-122; the
late-binding-lambda bug §8 B3 cannot occur here — multipliers are data, not
closures). One fused-kernel launch per group per step; fp32 optimizer state
and fp32 master weights when the model runs in bf16; per-submodel global-norm
gradient clipping with the cross-rank reduction done by the caller.
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch

from ..ops import grad_l2_norm_sq, multi_tensor_adamw_

logger = logging.getLogger("dinov3")


class FusedAdamW:
    def __init__(self, param_groups: List[dict], beta1: float = 0.9, beta2: float = 0.999,
                 eps: float = 1e-8, use_master_weights: bool = True):
        self.groups = param_groups
        self.beta1 = beta1
        self.beta2 = beta2
        self.eps = eps
        self.step_count = 0
        for g in self.groups:
            g["exp_avg"] = [torch.zeros_like(p, dtype=torch.float32) for p in g["params"]]
            g["exp_avg_sq"] = [torch.zeros_like(p, dtype=torch.float32) for p in g["params"]]
            if use_master_weights and g["params"] and g["params"][0].dtype != torch.float32:
                g["master"] = [p.detach().clone().float() for p in g["params"]]
            else:
                g["master"] = None

    @torch.no_grad()
    def grad_norms_per_submodel(self) -> Dict[str, torch.Tensor]:
        """Local sum-of-squares per submodel (caller all-reduces if sharded)."""
        sums: Dict[str, torch.Tensor] = {}
        for g in self.groups:
            grads = [p.grad for p in g["params"] if p.grad is not None]
            if not grads:
                continue
            s = grad_l2_norm_sq(grads)
            key = g["submodel"]
            sums[key] = sums.get(key, 0.0) + s
        return sums

    @torch.no_grad()
    def step(self, lr: float, weight_decay: float, last_layer_lr: Optional[float] = None,
             clip_scales: Optional[Dict[str, float]] = None) -> None:
        """clip_scales: per-submodel multiplicative grad scale (<=1) from the
        global-norm clip; last_layer_lr overrides lr for is_last_layer groups
        (the freeze window sets it to 0)."""
        self.step_count += 1
        for g in self.groups:
            params = [p for p in g["params"] if p.grad is not None]
            if not params:
                continue
            grads = [p.grad for p in params]
            if len(params) != len(g["params"]):
                # build matching state subsets (rare: frozen params)
                idx = [i for i, p in enumerate(g["params"]) if p.grad is not None]
                exp_avg = [g["exp_avg"][i] for i in idx]
                exp_avg_sq = [g["exp_avg_sq"][i] for i in idx]
                master = [g["master"][i] for i in idx] if g["master"] is not None else None
            else:
                exp_avg, exp_avg_sq, master = g["exp_avg"], g["exp_avg_sq"], g["master"]
            group_lr = last_layer_lr if (g["is_last_layer"] and last_layer_lr is not None) else lr
            group_lr *= g["lr_multiplier"]
            group_wd = weight_decay * g["wd_multiplier"]
            scale = 1.0
            if clip_scales is not None:
                scale = float(clip_scales.get(g["submodel"], 1.0))
            multi_tensor_adamw_(
                params, grads, exp_avg, exp_avg_sq, master,
                lr=group_lr, beta1=self.beta1, beta2=self.beta2, eps=self.eps,
                weight_decay=group_wd, step=self.step_count, grad_scale=scale,
            )

    def zero_grad(self, set_to_none: bool = True) -> None:
        for g in self.groups:
            for p in g["params"]:
                if set_to_none:
                    p.grad = None
                elif p.grad is not None:
                    p.grad.zero_()

    def state_dict(self) -> dict:
        return {
            "step_count": self.step_count,
            "groups": [
                {
                    "names": g["names"],
                    "exp_avg": g["exp_avg"],
                    "exp_avg_sq": g["exp_avg_sq"],
                    "master": g["master"],
                }
                for g in self.groups
            ],
        }

    def load_state_dict(self, state: dict) -> None:
        self.step_count = state["step_count"]
        by_names = {tuple(gs["names"]): gs for gs in state["groups"]}
        for g in self.groups:
            gs = by_names.get(tuple(g["names"]))
            if gs is None:
                logger.warning("optimizer group %s not found in checkpoint", g["names"][:1])
                continue
            for dst, src in zip(g["exp_avg"], gs["exp_avg"]):
                dst.copy_(src)
            for dst, src in zip(g["exp_avg_sq"], gs["exp_avg_sq"]):
                dst.copy_(src)
            if g["master"] is not None and gs["master"] is not None:
                for dst, src in zip(g["master"], gs["master"]):
                    dst.copy_(src)
