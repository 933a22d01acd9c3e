"""Pre-norm transformer block on the flat multi-crop token buffer, with
LayerScale and stochastic depth as batch-subset compute.

The whole residual stream lives in one flat [R, D] tensor; norms, qkv/proj
GEMMs and the FFN run on it directly. Stochastic depth gathers the kept
samples' rows (per crop group), computes on the smaller flat buffer, and
index_add's the rescaled residual back — the reference's throughput feature
(dinov3_jax/layers/block.py:88-194) without any per-block concat/split.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from ..ops.flat_attention import GroupMeta
from ..utils.utils import cat_keep_shapes, uncat_with_shapes
from .attention import RopeSinCos, SelfAttention
from .ffn_layers import Mlp
from .norms import LayerScale


class DropPathPlan:
    """Batched stochastic-depth bookkeeping for a whole forward pass.

    The per-sublayer `_subset_rows` costs a randperm (rand + radix sort) plus
    four index kernels per crop group — ~1000 tiny launches per ViT-L step.
    This plan draws EVERY sublayer's row subset up front: one rand + one
    batched argsort + one fused index build per crop group per step; each
    sublayer then just slices a row of the precomputed table. Stateless by
    sublayer index, so activation-checkpoint recompute replays identically.
    """

    def __init__(self, metas: List[GroupMeta], keep_ratio: float, n_sublayers: int, device):
        rows_parts = []
        scales = []
        new_metas: List[GroupMeta] = []
        new_off = 0
        for (off, B, N, sin, cos, prefix) in metas:
            keep = max(int(B * keep_ratio), 1)
            perm = torch.argsort(torch.rand(n_sublayers, B, device=device), dim=1)[:, :keep]
            r = off + perm.unsqueeze(2) * N + torch.arange(N, device=device)
            rows_parts.append(r.reshape(n_sublayers, keep * N))
            new_metas.append((new_off, keep, N, sin, cos, prefix))
            scales.append(torch.full((keep * N,), B / keep, device=device))
            new_off += keep * N
        self.rows = torch.cat(rows_parts, dim=1)  # [n_sublayers, total_keep_rows]
        self.scale = torch.cat(scales)
        self.metas = new_metas

    def take(self, sublayer: int) -> Tuple[torch.Tensor, List[GroupMeta], torch.Tensor]:
        return self.rows[sublayer], self.metas, self.scale


def _subset_rows(metas: List[GroupMeta], keep_ratio: float, device) -> Tuple[torch.Tensor, List[GroupMeta], torch.Tensor]:
    """Random per-group sample subsets -> (row indices, new metas, per-row scale)."""
    rows = []
    new_metas: List[GroupMeta] = []
    scales = []
    new_off = 0
    for (off, B, N, sin, cos, prefix) in metas:
        keep = max(int(B * keep_ratio), 1)
        idx = torch.randperm(B, device=device)[:keep]
        r = (off + idx.unsqueeze(1) * N + torch.arange(N, device=device).unsqueeze(0)).reshape(-1)
        rows.append(r)
        new_metas.append((new_off, keep, N, sin, cos, prefix))
        scales.append(torch.full((keep * N,), B / keep, device=device))
        new_off += keep * N
    return torch.cat(rows), new_metas, torch.cat(scales)


class SelfAttentionBlock(nn.Module):
    def __init__(
        self,
        dim: int,
        num_heads: int,
        ffn_ratio: float = 4.0,
        qkv_bias: bool = False,
        proj_bias: bool = True,
        ffn_bias: bool = True,
        drop: float = 0.0,
        attn_drop: float = 0.0,
        init_values: Optional[float] = None,
        drop_path: float = 0.0,
        norm_layer=None,
        ffn_layer=None,
        mask_k_bias: bool = False,
    ):
        super().__init__()
        from .norms import LayerNorm

        norm_layer = norm_layer or (lambda d: LayerNorm(d, eps=1e-6))
        ffn_layer = ffn_layer or Mlp
        self.norm1 = norm_layer(dim)
        self.attn = SelfAttention(
            dim, num_heads=num_heads, qkv_bias=qkv_bias, proj_bias=proj_bias,
            attn_drop=attn_drop, proj_drop=drop, mask_k_bias=mask_k_bias,
        )
        self.ls1 = LayerScale(dim, init_values) if init_values is not None else nn.Identity()
        self.norm2 = norm_layer(dim)
        self.mlp = ffn_layer(in_features=dim, hidden_features=int(dim * ffn_ratio), drop=drop, use_bias=ffn_bias)
        self.ls2 = LayerScale(dim, init_values) if init_values is not None else nn.Identity()
        self.sample_drop_ratio = drop_path

    def _add_scaled(self, flat: torch.Tensor, res: torch.Tensor, ls) -> torch.Tensor:
        """flat + LayerScale(res), fused on GPU when a gamma exists."""
        from ..ops.ls_axpy import ls_axpy

        if isinstance(ls, LayerScale):
            return ls_axpy(flat, res, ls.gamma)
        return flat + res

    def _fusable(self) -> Tuple[bool, bool]:
        """Which sublayers can fold (gamma, GEMM-bias) into the residual
        kernel (DINOV3_FUSED_RESIDUAL=1, ops/fused_residual.py)."""
        from ..ops.fused_residual import fused_residual_enabled

        if not fused_residual_enabled():
            return False, False
        fa = (isinstance(self.ls1, LayerScale) and self.attn.proj.bias is not None
              and not isinstance(self.attn.proj_drop, nn.Dropout))
        fm = (isinstance(self.ls2, LayerScale) and isinstance(self.mlp, Mlp)
              and self.mlp.fc2.bias is not None
              and not isinstance(self.mlp.drop, nn.Dropout))
        return fa, fm

    # ------------------------------------------------------------------
    def forward_flat(self, flat: torch.Tensor, metas: List[GroupMeta],
                     plan: Optional[DropPathPlan] = None, block_idx: int = 0,
                     inplace_ok: bool = True) -> torch.Tensor:
        fa, fm = self._fusable()
        if not (self.training and self.sample_drop_ratio > 0.0):
            if fa:
                from ..ops.fused_residual import ls_axpy_bias

                res = self.attn.forward_flat(self.norm1(flat), metas, skip_proj_bias=True)
                flat = ls_axpy_bias(flat, res, self.ls1.gamma, self.attn.proj.bias)
            else:
                flat = self._add_scaled(flat, self.attn.forward_flat(self.norm1(flat), metas), self.ls1)
            if fm:
                from ..ops.fused_residual import ls_axpy_bias

                res = self.mlp(self.norm2(flat), skip_out_bias=True)
                flat = ls_axpy_bias(flat, res, self.ls2.gamma, self.mlp.fc2.bias)
            else:
                flat = self._add_scaled(flat, self.mlp(self.norm2(flat)), self.ls2)
            return flat

        from ..ops.fused_residual import ls_scatter_add_rows
        from ..ops.row_ops import gather_rows, scatter_add_rows

        keep_ratio = 1.0 - self.sample_drop_ratio
        if plan is not None:
            rows1, metas1, scale1 = plan.take(2 * block_idx)
        else:
            rows1, metas1, scale1 = _subset_rows(metas, keep_ratio, flat.device)
        sub = gather_rows(flat, rows1)
        if fa:
            if not inplace_ok:
                flat = flat.clone()
            res = self.attn.forward_flat(self.norm1(sub), metas1, skip_proj_bias=True)
            flat = ls_scatter_add_rows(flat, rows1, res, self.ls1.gamma,
                                       self.attn.proj.bias, scale1)
        else:
            res = self.ls1(self.attn.forward_flat(self.norm1(sub), metas1))
            flat = scatter_add_rows(flat, rows1, res, scale1)

        if plan is not None:
            rows2, _, scale2 = plan.take(2 * block_idx + 1)
        else:
            rows2, _, scale2 = _subset_rows(metas, keep_ratio, flat.device)
        sub = gather_rows(flat, rows2)
        if fm:
            res = self.mlp(self.norm2(sub), skip_out_bias=True)
            return ls_scatter_add_rows(flat, rows2, res, self.ls2.gamma,
                                       self.mlp.fc2.bias, scale2)
        res = self.ls2(self.mlp(self.norm2(sub)))
        return scatter_add_rows(flat, rows2, res, scale2)

    # ------------------------------------------------------------------
    def forward(self, x: torch.Tensor, rope: Optional[RopeSinCos] = None) -> torch.Tensor:
        return self.forward_list([x], [rope])[0]

    def forward_list(self, x_list: List[torch.Tensor],
                     rope_list: Optional[List[Optional[RopeSinCos]]] = None) -> List[torch.Tensor]:
        if rope_list is None:
            rope_list = [None] * len(x_list)
        flat, shapes, counts = cat_keep_shapes(x_list)
        metas = []
        off = 0
        for x, rope in zip(x_list, rope_list):
            metas.append(SelfAttention._meta_for(x, rope, off))
            off += x.shape[0] * x.shape[1]
        out = self.forward_flat(flat, metas)
        return uncat_with_shapes(out, shapes, counts)
