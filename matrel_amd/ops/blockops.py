"""Per-block math kernels (dispatch layer).

Reference equivalent: MLMatrix/LocalMatrix per-block algebra backed by
Breeze/netlib BLAS (SURVEY.md §2.1 "Per-block math kernels"). Here every
GPU-side op dispatches to the hand-written CDNA4 HIP extension
(matrel_amd/_C); the CPU path is plain torch and exists for tests and the
no-GPU config-1 slice.

Conventions:
  * Dense blocks are 2D contiguous row-major torch tensors (bf16 or fp32
    on GPU; fp32 on CPU).
  * GEMM accumulates in fp32. The HIP GEMM takes B TRANSPOSED
    (C += A @ Bt^T) so both operands stage identically into LDS; callers
    that reuse a B block across many A blocks pre-transpose once via
    `transpose()` and call `gemm_bt_acc` (summa.py, local_matmul).
  * Kernel tiles need M,N % 128 == 0 and K % 64 (bf16) / % 32 (f32):
    `_pad2` pads ragged blocks transparently.
"""

from __future__ import annotations

from typing import Optional

import torch

from .backend import require_ext, use_hip

# elementwise op ids shared with csrc/elementwise.hip
EW_ADD, EW_SUB, EW_MUL, EW_DIV, EW_MIN, EW_MAX = 0, 1, 2, 3, 4, 5
# scalar op ids
SC_ADD, SC_MUL, SC_POW = 0, 1, 2
# reduction kinds shared with csrc/reduce.hip
RED_SUM, RED_MAX, RED_MIN, RED_NNZ, RED_SQSUM = 0, 1, 2, 3, 4

_EW_TORCH = {
    EW_ADD: torch.add, EW_SUB: torch.sub,
    EW_MUL: torch.mul, EW_DIV: torch.div,
    EW_MIN: torch.minimum, EW_MAX: torch.maximum,
}

_TILE_MN = 128


def is_sparse(t: torch.Tensor) -> bool:
    return t.layout == torch.sparse_csr


def densify(t: torch.Tensor) -> torch.Tensor:
    return t.to_dense() if is_sparse(t) else t


def maybe_sparsify(t: torch.Tensor, threshold: float) -> torch.Tensor:
    """Store a block CSR when its density is below `threshold` (the
    reference's per-block auto dense/sparse selection by nnz,
    SURVEY.md §2.1 "Block-matrix representation")."""
    if is_sparse(t) or threshold <= 0:
        return t
    nnz = int((t != 0).sum().item())
    if t.numel() and nnz / t.numel() < threshold:
        return t.to_sparse_csr()
    return t


def spmm_acc(acc: torch.Tensor, a_csr: torch.Tensor,
             b: torch.Tensor) -> None:
    """acc(fp32, MxN) += a_csr(M x K) @ b(K x N) - within-block sparse
    product (HIP CSR kernel on GPU)."""
    if acc.is_cuda:
        ext = require_ext()
        crow = a_csr.crow_indices().to(torch.int32)
        cols = a_csr.col_indices().to(torch.int32)
        vals = a_csr.values()
        if vals.dtype != b.dtype:
            vals, b = vals.float(), b.float()
        ext.spmm_csr_acc(acc, crow, cols, vals.contiguous(),
                         b.contiguous())
    else:
        acc.add_(a_csr.to_dense().float() @ b.float())


def spgemm_acc(acc: torch.Tensor, a_csr: torch.Tensor,
               b_csr: torch.Tensor) -> bool:
    """acc(fp32, MxN) += a_csr(M x K) @ b_csr(K x N) - sparse x sparse
    block product (HIP ds_add_f32 LDS-row kernel on GPU; SURVEY.md §2.1
    "sparse x sparse block products"). Returns False when the kernel
    declined (N > 16 KiB LDS row) - caller densifies b instead.

    The kernel accumulates via LDS float ATOMICS (lane-order varies), so
    in deterministic mode (utils/debug.py: fixed-order reductions only)
    it declines and the caller takes the deterministic densified path."""
    from ..config import get_config
    if acc.is_cuda and get_config().deterministic:
        return False
    if acc.is_cuda:
        ext = require_ext()
        va = a_csr.values()
        vb = b_csr.values()
        if va.dtype != vb.dtype:
            va, vb = va.float(), vb.float()
        return ext.spgemm_csr_csr_acc(
            acc, a_csr.crow_indices().to(torch.int32),
            a_csr.col_indices().to(torch.int32), va.contiguous(),
            b_csr.crow_indices().to(torch.int32),
            b_csr.col_indices().to(torch.int32), vb.contiguous())
    d = a_csr.to_dense().float() @ b_csr.to_dense().float()
    # ragged blocks: the product can be narrower/shorter than the
    # (block-size-wide) accumulator slice - missing columns/rows are
    # zero contributions (found by scripts/cpu_fuzz_sweep.py)
    acc[: d.shape[0], : d.shape[1]].add_(d)
    return True


def _kmult(dtype: torch.dtype) -> int:
    return 64 if dtype == torch.bfloat16 else 32


def _ceil(x: int, m: int) -> int:
    return (x + m - 1) // m * m


def _pad2(t: torch.Tensor, rows: int, cols: int) -> torch.Tensor:
    if t.shape[0] == rows and t.shape[1] == cols:
        return t.contiguous()
    out = torch.zeros(rows, cols, dtype=t.dtype, device=t.device)
    out[: t.shape[0], : t.shape[1]] = t
    return out


def gemm_bt_acc(acc: torch.Tensor, a: torch.Tensor, bt: torch.Tensor,
                beta0: bool = False) -> None:
    """acc(fp32, MxN) += a(MxK) @ bt(NxK)^T - the hot MFMA kernel.
    With beta0=True, acc is OVERWRITTEN (may be torch.empty): the kernel
    skips the C-read pass (first k-contribution of a block)."""
    if not use_hip(a):
        prod = a.float() @ bt.float().t()
        if beta0:
            acc.copy_(prod)
        else:
            acc.add_(prod)
        return
    ext = require_ext()
    if a.dtype != bt.dtype:
        a, bt = a.float(), bt.float()
    m, k = a.shape
    n = bt.shape[0]
    km = _kmult(a.dtype)
    mp, np_, kp = _ceil(m, _TILE_MN), _ceil(n, _TILE_MN), _ceil(k, km)
    if (mp, np_, kp) == (m, n, k):
        if beta0:
            ext.gemm_bt_acc_beta0(acc, a.contiguous(), bt.contiguous())
        else:
            ext.gemm_bt_acc(acc, a.contiguous(), bt.contiguous())
        return
    ap = _pad2(a, mp, kp)
    btp = _pad2(bt, np_, kp)
    cp = torch.empty(mp, np_, dtype=torch.float32, device=a.device)
    ext.gemm_bt_acc_beta0(cp, ap, btp)
    if beta0:
        acc.copy_(cp[:m, :n])
    else:
        acc.add_(cp[:m, :n])


def gemm_bt_final(acc: torch.Tensor, out_bf16: torch.Tensor,
                  a: torch.Tensor, bt: torch.Tensor,
                  beta0: bool = False) -> bool:
    """Fused last-k call: (acc-partials [+ a @ bt^T]) written STRAIGHT to
    bf16 `out_bf16` (no fp32 store, no conversion pass). Returns False
    when the geometry/dtype is unsupported - caller falls back to
    gemm_bt_acc + .to(bf16)."""
    if not use_hip(a) or a.dtype != torch.bfloat16 \
            or bt.dtype != torch.bfloat16 \
            or out_bf16.dtype != torch.bfloat16:
        return False
    m, k = a.shape
    n = bt.shape[0]
    if m % 128 or n % 128 or k % 64:
        return False
    return bool(require_ext().gemm_bt_final(
        acc, out_bf16, a.contiguous(), bt.contiguous(), beta0))


def gemm_acc(acc: torch.Tensor, a: torch.Tensor, b: torch.Tensor) -> None:
    """acc(fp32) += a @ b (transposes b internally on GPU; prefer
    gemm_bt_acc + a cached transpose when b is reused)."""
    if use_hip(a):
        gemm_bt_acc(acc, a, transpose(b))
    else:
        acc.add_(a.float() @ b.float())


def gemm(a: torch.Tensor, b: torch.Tensor,
         out_dtype: Optional[torch.dtype] = None) -> torch.Tensor:
    acc = torch.zeros(a.shape[0], b.shape[1], dtype=torch.float32,
                      device=a.device)
    gemm_acc(acc, a, b)
    return acc.to(out_dtype) if out_dtype is not None else acc


def ewise(op: int, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    a, b = densify(a), densify(b)
    if use_hip(a):
        if a.dtype != b.dtype:
            a, b = a.float(), b.float()
        return require_ext().ewise(op, a.contiguous(), b.contiguous())
    return _EW_TORCH[op](a, b)


def ewise_scalar(op: int, a: torch.Tensor, s: float) -> torch.Tensor:
    a = densify(a)
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
    """Materialized transpose (tiled LDS kernel on GPU)."""
    a = densify(a)
    if use_hip(a):
        return require_ext().transpose(a.contiguous())
    return a.t().contiguous()


def ewise_transpose(op: int, a: torch.Tensor,
                    b: torch.Tensor) -> torch.Tensor:
    """(a OP b)^T in ONE pass (fused kernel): 3 HBM
    streams instead of the 5 an ewise + transpose pair pays."""
    a, b = densify(a), densify(b)
    if use_hip(a):
        if a.dtype != b.dtype:
            a, b = a.float(), b.float()
        return require_ext().ewise_transpose(op, a.contiguous(),
                                             b.contiguous())
    return _EW_TORCH[op](a, b).t().contiguous()


def reduce2d(a: torch.Tensor, kind: int, axis: Optional[int]
             ) -> torch.Tensor:
    """Reduce a block. axis=None -> 1-elem tensor, axis=0 -> per-column,
    axis=1 -> per-row. fp32 output."""
    if is_sparse(a):
        if axis is None and kind in (RED_SUM, RED_NNZ, RED_SQSUM):
            # reduce over the values vector directly (zeros contribute 0)
            v = a.values().reshape(1, -1)
            if v.numel() == 0:
                return torch.zeros(1, dtype=torch.float32, device=a.device)
            return reduce2d(v.contiguous(), kind, None)
        a = a.to_dense()
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
    elif kind == RED_SQSUM:
        sq = af * af
        r = sq.sum() if axis is None else sq.sum(dim=axis)
    else:
        raise ValueError(kind)
    return r.reshape(-1) if axis is None else r


def trace_block(a: torch.Tensor) -> torch.Tensor:
    """Sum of the main diagonal (1-elem fp32 tensor)."""
    a = densify(a)
    if use_hip(a):
        return require_ext().trace(a.contiguous())
    return torch.diagonal(a.float()).sum().reshape(1)


def outer(op: str, u: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """Cross-join outer op: out[i,j] = u[i] * v[j] (op='mul') or + ."""
    if use_hip(u):
        return require_ext().outer(0 if op == "mul" else 1,
                                   u.contiguous(), v.contiguous())
    uf, vf = u.float().reshape(-1), v.float().reshape(-1)
    if op == "mul":
        return torch.outer(uf, vf)
    return uf.unsqueeze(1) + vf.unsqueeze(0)


def mask_rows_cols(a: torch.Tensor, row_mask: Optional[torch.Tensor],
                   col_mask: Optional[torch.Tensor]) -> torch.Tensor:
    """Zero out rows/cols excluded by boolean masks (boundary blocks of an
    index-predicate selection; fully excluded blocks are pruned at plan
    time and never reach a kernel - SURVEY.md §3.4)."""
    a = densify(a)
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
    a = densify(a)
    if use_hip(a):
        return require_ext().mask_value(a.contiguous(), float(lo), float(hi),
                                        bool(keep_inside))
    inside = (a >= lo) & (a <= hi)
    keep = inside if keep_inside else ~inside
    return torch.where(keep, a, torch.zeros((), dtype=a.dtype,
                                            device=a.device))


def mask_isin(a: torch.Tensor, needles_sorted: torch.Tensor) -> torch.Tensor:
    """Keep entries of `a` whose value is in the sorted needle set."""
    a = densify(a)
    if use_hip(a):
        return require_ext().mask_isin(a.contiguous(), needles_sorted)
    keep = torch.isin(a.float(), needles_sorted.float())
    return torch.where(keep, a, torch.zeros((), dtype=a.dtype,
                                            device=a.device))


def to_mx8(t_bf16: torch.Tensor):
    """bf16 -> (e4m3 bytes, e8m0 block scales [rows][cols/32])."""
    return tuple(require_ext().mx8_convert(t_bf16.contiguous()))


def to_mx4(t_bf16: torch.Tensor):
    """bf16 -> (packed e2m1 nibbles [rows][cols/2], packed e8m0
    scales)."""
    return tuple(require_ext().mx4_convert(t_bf16.contiguous()))


def to_mx6(t_bf16: torch.Tensor):
    """bf16 -> (packed e2m3 6-bit fields [rows][cols*3/4], packed e8m0
    scales)."""
    return tuple(require_ext().mx6_convert(t_bf16.contiguous()))


def gemm_bt_mx6_acc(acc: torch.Tensor, a6, asc, bt6, btsc,
                    beta0: bool = False) -> None:
    """acc(fp32) += dequant(a6,asc) @ dequant(bt6,btsc)^T via the MX-fp6
    scaled MFMA (e2m3: fp8-class accuracy at the fp4 rate)."""
    require_ext().gemm_bt_mx6_acc(acc, a6, asc, bt6, btsc, beta0)


def gemm_bt_mx4_acc(acc: torch.Tensor, a4, asc, bt4, btsc,
                    beta0: bool = False) -> None:
    """acc(fp32) += dequant(a4,asc) @ dequant(bt4,btsc)^T via the MX-fp4
    scaled MFMA (~2x the fp8 rate at ~1-2 significant digits)."""
    require_ext().gemm_bt_mx4_acc(acc, a4, asc, bt4, btsc, beta0)


def gemm_bt_mx8_acc(acc: torch.Tensor, a8, asc, bt8, btsc,
                    beta0: bool = False) -> None:
    """acc(fp32) += dequant(a8,asc) @ dequant(bt8,btsc)^T via the MX-fp8
    scaled MFMA (~2x the bf16 rate). Shapes must satisfy M,N % 256 == 0,
    K % 64 == 0 (callers guard and fall back to bf16)."""
    require_ext().gemm_bt_mx8_acc(acc, a8, asc, bt8, btsc, beta0)
