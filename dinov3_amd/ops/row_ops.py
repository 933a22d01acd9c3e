"""Row gather / scaled scatter-add for stochastic-depth subset compute (K11)."""

from __future__ import annotations

import torch

_EMPTY_F = {}


def _empty_f(device):
    key = str(device)
    if key not in _EMPTY_F:
        _EMPTY_F[key] = torch.empty(0, dtype=torch.float32, device=device)
    return _EMPTY_F[key]


class _GatherRowsFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, flat, idx, out_rows):
        from . import hip_ops

        ctx.save_for_backward(idx)
        ctx.src_rows = flat.shape[0]
        return hip_ops().row_gather(flat, idx)

    @staticmethod
    def backward(ctx, dout):
        from . import hip_ops

        (idx,) = ctx.saved_tensors
        dflat = torch.zeros(ctx.src_rows, dout.shape[-1], dtype=dout.dtype,
                            device=dout.device)
        hip_ops().row_scatter_add_(dflat, idx, dout.contiguous(), _empty_f(dout.device))
        return dflat, None, None


class _ScatterAddRowsFn(torch.autograd.Function):
    """out = flat with out[idx] += res * scale (rows disjoint)."""

    @staticmethod
    def forward(ctx, flat, idx, res, scale):
        from . import hip_ops

        out = flat.clone()
        hip_ops().row_scatter_add_(out, idx, res.contiguous(), scale)
        ctx.save_for_backward(idx, scale)
        return out

    @staticmethod
    def backward(ctx, dout):
        from . import hip_ops

        idx, scale = ctx.saved_tensors
        dout = dout.contiguous()
        dres = hip_ops().row_gather_scaled(dout, idx, scale)
        return dout, None, dres, None


def gather_rows(flat: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    from . import use_hip

    if use_hip(flat) and flat.shape[-1] % 8 == 0:
        return _GatherRowsFn.apply(flat, idx, None)
    return flat.index_select(0, idx)


def scatter_add_rows(flat: torch.Tensor, idx: torch.Tensor, res: torch.Tensor,
                     scale: torch.Tensor) -> torch.Tensor:
    """flat.index_add(0, idx, res * scale[:, None]) with fused scaling."""
    from . import use_hip

    if use_hip(flat) and flat.shape[-1] % 8 == 0:
        return _ScatterAddRowsFn.apply(flat, idx, res, scale.float().contiguous())
    return flat.index_add(0, idx, (res * scale.unsqueeze(1)).to(flat.dtype))
