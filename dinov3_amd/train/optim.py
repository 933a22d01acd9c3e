"""FusedAdamW: single-launch multi-tensor AdamW over fused param groups
(SURVEY K24/K25).

Replaces the reference's optax.multi_transform tower (train.py:75-122; the
late-binding-lambda bug §8 B3 cannot occur here — multipliers are data). On
GPU the whole optimizer step is ONE kernel launch over a cached device-side
plan (per-tensor lr/wd multipliers, last-layer flag, submodel id); the
per-submodel global-norm clip factors stay on device end to end (no host
sync in the step). fp32 optimizer state + fp32 master weights for bf16
params.
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Union

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
        self.submodels = sorted({g["submodel"] for g in self.groups})
        self._sub_idx = {name: i for i, name in enumerate(self.submodels)}
        for g in self.groups:
            g["exp_avg"] = [torch.zeros_like(p, dtype=torch.float32) for p in g["params"]]
            g["exp_avg_sq"] = [torch.zeros_like(p, dtype=torch.float32) for p in g["params"]]
            if use_master_weights and g["params"] and g["params"][0].dtype != torch.float32:
                g["master"] = [p.detach().clone().float() for p in g["params"]]
            else:
                g["master"] = None
        self._plan = None
        self._has_master = any(g["master"] is not None for g in self.groups)

    # ------------------------------------------------------------------
    def _ensure_plan(self):
        """Build the single-launch plan (GPU only, needs grads materialized)."""
        if self._plan is not None:
            return self._plan
        from ..ops.mt_plan import MultiTensorPlan

        p_list, g_list, m_list, v_list, w_list = [], [], [], [], []
        lr_mult, wd_mult, is_last, sub_id = [], [], [], []
        for g in self.groups:
            for i, p in enumerate(g["params"]):
                if p.grad is None:
                    # params a config never trains (e.g. the ibot heads with
                    # ibot.loss_weight=0) get a persistent zero grad so the
                    # single-launch plan holds a pointer for them; autograd
                    # accumulates into it if they ever receive gradients
                    p.grad = torch.zeros_like(p)
                p_list.append(p)
                g_list.append(p.grad)
                m_list.append(g["exp_avg"][i])
                v_list.append(g["exp_avg_sq"][i])
                if g["master"] is not None:
                    w_list.append(g["master"][i])
                lr_mult.append(g["lr_multiplier"])
                wd_mult.append(g["wd_multiplier"])
                is_last.append(1.0 if g["is_last_layer"] else 0.0)
                sub_id.append(self._sub_idx[g["submodel"]])
        lists = [p_list, g_list, m_list, v_list]
        if w_list:
            assert len(w_list) == len(p_list), "mixed master/non-master groups unsupported"
            lists.append(w_list)
        device = p_list[0].device
        self._plan = MultiTensorPlan(lists)
        self._lr_mult = torch.tensor(lr_mult, dtype=torch.float32, device=device)
        self._wd_mult = torch.tensor(wd_mult, dtype=torch.float32, device=device)
        self._is_last = torch.tensor(is_last, dtype=torch.float32, device=device)
        self._sub_id = torch.tensor(sub_id, dtype=torch.int32, device=device)
        logger.info("fused optimizer plan: %d tensors, %d submodels", len(p_list),
                    len(self.submodels))
        return self._plan

    def _use_planned(self) -> bool:
        return bool(self.groups) and self.groups[0]["params"][0].is_cuda

    # ------------------------------------------------------------------
    # Grads are fully replicated on every rank after GradReducer.finalize, so
    # the per-rank sums are already global — the trainer must NOT all-reduce
    # them (that would inflate squared norms by world_size).
    needs_norm_allreduce = False

    @torch.no_grad()
    def grad_norm_sums(self) -> torch.Tensor:
        """Sum of squared grads per submodel — [n_submodels] fp32 on the
        params' device (already global in the replicated-grad DDP path)."""
        if self._use_planned():
            from ..ops.mt_plan import l2norm_planned

            plan = self._ensure_plan()
            return l2norm_planned(plan, self._sub_id, len(self.submodels), list_index=1)
        sums = torch.zeros(len(self.submodels), dtype=torch.float32)
        for g in self.groups:
            grads = [p.grad for p in g["params"] if p.grad is not None]
            if grads:
                sums[self._sub_idx[g["submodel"]]] += grad_l2_norm_sq(grads)
        return sums

    def clip_factors(self, sums: torch.Tensor, clip: float) -> torch.Tensor:
        """Per-submodel multiplicative grad scale from summed squared norms."""
        return torch.clamp(clip / (sums.sqrt() + 1e-6), max=1.0)

    # ------------------------------------------------------------------
    @torch.no_grad()
    def step(self, lr: float, weight_decay: float, last_layer_lr: Optional[float] = None,
             clip_scales: Union[None, Dict[str, float], torch.Tensor] = None) -> None:
        self.step_count += 1
        if last_layer_lr is None:
            last_layer_lr = lr
        if self._use_planned():
            from ..ops.mt_plan import adamw_planned

            plan = self._ensure_plan()
            if clip_scales is None:
                clip_t = torch.ones(len(self.submodels), dtype=torch.float32,
                                    device=plan.device)
            elif isinstance(clip_scales, dict):
                clip_t = torch.tensor(
                    [clip_scales.get(s, 1.0) for s in self.submodels],
                    dtype=torch.float32, device=plan.device,
                )
            else:
                clip_t = clip_scales.float()
            adamw_planned(plan, self._lr_mult, self._wd_mult, self._is_last, self._sub_id,
                          clip_t, lr, last_layer_lr, weight_decay, self.beta1, self.beta2,
                          self.eps, self.step_count, self._has_master)
            return
        # CPU / per-group fallback
        for g in self.groups:
            params = [p for p in g["params"] if p.grad is not None]
            if not params:
                continue
            grads = [p.grad for p in params]
            if len(params) != len(g["params"]):
                idx = [i for i, p in enumerate(g["params"]) if p.grad is not None]
                exp_avg = [g["exp_avg"][i] for i in idx]
                exp_avg_sq = [g["exp_avg_sq"][i] for i in idx]
                master = [g["master"][i] for i in idx] if g["master"] is not None else None
            else:
                exp_avg, exp_avg_sq, master = g["exp_avg"], g["exp_avg_sq"], g["master"]
            group_lr = last_layer_lr if g["is_last_layer"] else lr
            group_lr *= g["lr_multiplier"]
            group_wd = weight_decay * g["wd_multiplier"]
            scale = 1.0
            if clip_scales is not None:
                if isinstance(clip_scales, dict):
                    scale = float(clip_scales.get(g["submodel"], 1.0))
                else:
                    scale = float(clip_scales[self._sub_idx[g["submodel"]]])
            multi_tensor_adamw_(
                params, grads, exp_avg, exp_avg_sq, master,
                lr=group_lr, beta1=self.beta1, beta2=self.beta2, eps=self.eps,
                weight_decay=group_wd, step=self.step_count, grad_scale=scale,
            )

    def zero_grad(self, set_to_none: Optional[bool] = None) -> None:
        # once a plan exists, grads must keep their storage: zero in place
        # (one foreach launch, not one fill per tensor)
        if set_to_none is None:
            set_to_none = self._plan is None
        if set_to_none:
            for g in self.groups:
                for p in g["params"]:
                    p.grad = None
            return
        grads = [p.grad for g in self.groups for p in g["params"] if p.grad is not None]
        if grads:
            torch._foreach_zero_(grads)

    # ------------------------------------------------------------------
    state_format = "groups"

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

    def load_canonical(self, canon: dict) -> None:
        """Restore from the world-size-independent canonical form
        (optim_state.py): split each group's flat fp32 vectors back into
        per-param tensors."""
        self.step_count = canon["step_count"]
        for g in self.groups:
            gs = canon["groups"].get(tuple(g["names"]))
            if gs is None:
                logger.warning("optimizer group %s missing from canonical state", g["names"][:1])
                continue
            for field, dsts in (("exp_avg", g["exp_avg"]), ("exp_avg_sq", g["exp_avg_sq"]),
                                ("master", g["master"])):
                src = gs.get(field)
                if dsts is None or src is None:
                    continue
                off = 0
                for dst in dsts:
                    n = dst.numel()
                    dst.copy_(src[off: off + n].view(dst.shape).to(dst.device))
                    off += n
