"""Per-block math kernels (dispatch layer).

Reference equivalent: MLMatrix/LocalMatrix per-block algebra backed by
Breeze/netlib BLAS (SURVEY.md §2.1 "Per-block math kernels"). Here every
GPU-side op dispatches to the hand-written CDNA4 HIP extension
(matrel_amd/_C); the CPU path is plain torch and exists for tests and the
no-GPU config-1 slice.

Conventions:
  * Dense blocks are 2D contiguous row-major torch tensors.
  * GEMM accumulates in fp32 (`acc` tensors are fp32); inputs bf16 on GPU.
  * Elementwise ops are fused single-pass kernels on GPU (HBM-bound).
"""

from __future__ import annotations

from typing import Optional

import torch

from .backend import require_ext, use_hip

# elementwise op ids shared with csrc/elementwise.hip
EW_ADD, EW_SUB, EW_MUL, EW_DIV = 0, 1, 2, 3
# scalar op ids
SC_ADD, SC_MUL, SC_POW = 0, 1, 2
# reduction kinds
RED_SUM, RED_MAX, RED_MIN, RED_NNZ = 0, 1, 2, 3

_EW_TORCH = {
    EW_ADD: torch.add, EW_SUB: torch.sub,
    EW_MUL: torch.mul, EW_DIV: torch.div,
}


def gemm_acc(acc: torch.Tensor, a: torch.Tensor, b: torch.Tensor) -> None:
    """acc(fp32) += a @ b. The hot kernel (MFMA bf16 on GPU)."""
    if use_hip(a):
        require_ext().gemm_bf16_acc(acc, a.contiguous(), b.contiguous())
    else:
        acc.add_(a.float() @ b.float())


def gemm(a: torch.Tensor, b: torch.Tensor,
         out_dtype: Optional[torch.dtype] = None) -> torch.Tensor:
    acc = torch.zeros(a.shape[0], b.shape[1], dtype=torch.float32,
                      device=a.device)
    gemm_acc(acc, a, b)
    return acc.to(out_dtype) if out_dtype is not None else acc


def ewise(op: int, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    if use_hip(a):
        return require_ext().ewise(op, a.contiguous(), b.contiguous())
    return _EW_TORCH[op](a, b)


def ewise_scalar(op: int, a: torch.Tensor, s: float) -> torch.Tensor:
    if use_hip(a):
        return require_ext().ewise_scalar(op, a.contiguous(), float(s))
    if op == SC_ADD:
        return a + s
    if op == SC_MUL:
        return a * s
    if op == SC_POW:
        return a.float().pow(s).to(a.dtype)
    raise ValueError(op)


def transpose(a: torch.Tensor) -> torch.Tensor:
    """Materialized transpose (fused tiled kernel on GPU)."""
    if use_hip(a):
        return require_ext().transpose(a.contiguous())
    return a.t().contiguous()


def reduce2d(a: torch.Tensor, kind: int, axis: Optional[int]
             ) -> torch.Tensor:
    """Reduce a block. axis=None -> scalar (0-dim is returned as 1-elem),
    axis=0 -> per-column vector, axis=1 -> per-row vector. fp32 output."""
    if use_hip(a):
        return require_ext().reduce2d(a.contiguous(), kind,
                                      -1 if axis is None else axis)
    af = a.float()
    if kind == RED_SUM:
        r = af.sum() if axis is None else af.sum(dim=axis)
    elif kind == RED_MAX:
        r = af.max() if axis is None else af.max(dim=axis).values
    elif kind == RED_MIN:
        r = af.min() if axis is None else af.min(dim=axis).values
    elif kind == RED_NNZ:
        nz = (af != 0).float()
        r = nz.sum() if axis is None else nz.sum(dim=axis)
    else:
        raise ValueError(kind)
    return r.reshape(-1) if axis is None else r


def mask_rows_cols(a: torch.Tensor, row_mask: Optional[torch.Tensor],
                   col_mask: Optional[torch.Tensor]) -> torch.Tensor:
    """Zero out rows/cols excluded by boolean masks (boundary blocks of an
    index-predicate selection; fully excluded blocks are pruned at plan
    time and never reach a kernel - SURVEY.md §3.4)."""
    if use_hip(a):
        e = require_ext()
        rm = row_mask.to(a.device) if row_mask is not None else None
        cm = col_mask.to(a.device) if col_mask is not None else None
        return e.mask_rows_cols(a.contiguous(), rm, cm)
    out = a
    if row_mask is not None:
        out = out * row_mask.to(out.dtype).unsqueeze(1)
    if col_mask is not None:
        out = out * col_mask.to(out.dtype).unsqueeze(0)
    return out


def mask_value(a: torch.Tensor, lo: float, hi: float,
               keep_inside: bool) -> torch.Tensor:
    """Value-predicate selection: keep entries inside [lo, hi] (or outside),
    zeroing the rest. Fused single-pass on GPU."""
    if use_hip(a):
        return require_ext().mask_value(a.contiguous(), float(lo), float(hi),
                                        bool(keep_inside))
    inside = (a >= lo) & (a <= hi)
    keep = inside if keep_inside else ~inside
    return torch.where(keep, a, torch.zeros((), dtype=a.dtype,
                                            device=a.device))
