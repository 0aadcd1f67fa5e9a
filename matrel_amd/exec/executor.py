"""DAG executor: evaluates an optimized logical plan + physical annotations.

Reference equivalent: Spark's DAGScheduler + the matrix extension's
physical executions. Here execution is a
host-side recursive walk (the DAG is small; per-node work is huge GPU
kernels / collectives), with:

  * memoization per node object - CSE-shared subtrees execute once
    (SURVEY.md §2.1 "reuse of common subexpressions"),
  * scheme repartitioning inserted where the physical plan demands it,
  * per-op tracing hooks (exec/tracing.py) for the chrome-trace output.

Aggregates return 1x1 / vector BlockMatrix results replicated on every
rank (BroadcastScheme); `Session.compute()` unwraps them to floats /
tensors.
"""

from __future__ import annotations

import math
from typing import Dict, Optional

import torch
import torch.distributed as dist

from ..blocks.matrix import BlockMatrix
from ..blocks.schemes import BCAST, COL, GRID, ROW
from ..ops import blockops
from ..plan.logical import (Agg, CrossJoin, Ewise, Inverse, Leaf, MatMul,
                            Plan, ScalarOp, Select, SemiJoinValue, Solve,
                            Transpose)
from ..plan.physical import PhysicalPlan, plan_physical
from .runtime import get_runtime
from .metrics import record_op
from .summa import local_matmul, summa_matmul
from .tracing import trace_op


class Executor:
    def __init__(self, physical: Optional[PhysicalPlan] = None) -> None:
        self.physical = physical
        self.cache: Dict[int, BlockMatrix] = {}

    # ------------------------------------------------------------------

    def execute(self, root: Plan) -> BlockMatrix:
        if self.physical is None:
            self.physical = plan_physical(root)
        return self._eval(root)

    def _eval(self, n: Plan) -> BlockMatrix:
        if id(n) in self.cache:
            return self.cache[id(n)]
        pi = self.physical.get(n)
        with trace_op(type(n).__name__, shape=n.shape), \
                record_op(type(n).__name__, pi.est_flops,
                          pi.est_comm_bytes):
            out = self._dispatch(n)
        tgt = self.physical.get(n).scheme
        if out.scheme != tgt and get_runtime().is_distributed:
            out = out.repartition(tgt)
        self.cache[id(n)] = out
        return out

    def _dispatch(self, n: Plan) -> BlockMatrix:
        if isinstance(n, Leaf):
            return n.data
        if isinstance(n, Transpose):
            return self._transpose(n)
        if isinstance(n, ScalarOp):
            return self._scalar(n)
        if isinstance(n, Ewise):
            return self._ewise(n)
        if isinstance(n, MatMul):
            return self._matmul(n)
        if isinstance(n, Select):
            return self._select(n)
        if isinstance(n, Agg):
            return self._agg(n)
        if isinstance(n, Solve):
            return self._solve(n)
        if isinstance(n, Inverse):
            return self._inverse(n)
        if isinstance(n, CrossJoin):
            return self._crossjoin(n)
        if isinstance(n, SemiJoinValue):
            return self._semijoin_value(n)
        raise TypeError(f"cannot execute {type(n).__name__}")

    # --- operators ------------------------------------------------------

    def _transpose(self, n: Transpose) -> BlockMatrix:
        # Fused (A op B)^T: single-pass ewise+transpose kernel per block
        #  - 3 HBM streams instead of 5. Applies when
        # the child is an un-memoized Ewise; a CSE-shared child that is
        # evaluated later elsewhere simply recomputes the (cheap) ewise.
        ch = n.children[0]
        locals_t: dict
        if (isinstance(ch, Ewise) and id(ch) not in self.cache
                and ch.op in ("add", "sub", "mul", "div", "min", "max")):
            a = self._eval(ch.children[0])
            b = self._eval(ch.children[1])
            if a.block_size != b.block_size:
                raise ValueError("elementwise op needs equal block sizes")
            tgt = self.physical.get(ch).scheme
            if a.scheme != tgt:
                a = a.repartition(tgt)
            if b.scheme != tgt:
                b = b.repartition(tgt)
            op = {"add": blockops.EW_ADD, "sub": blockops.EW_SUB,
                  "mul": blockops.EW_MUL, "div": blockops.EW_DIV,
                  "min": blockops.EW_MIN, "max": blockops.EW_MAX}[ch.op]
            if ch.op in ("add", "sub", "min", "max"):
                keys = set(a.blocks) | set(b.blocks)
                pm = a.presence() | b.presence()
            elif ch.op == "mul":
                keys = set(a.blocks) & set(b.blocks)
                pm = a.presence() & b.presence()
            else:  # div (matches _ewise's key rule)
                keys = set(a.blocks)
                pm = a.presence().clone()
            locals_t = {}
            for (bi, bj) in keys:
                ta = a.get_block(bi, bj)
                tb = b.get_block(bi, bj)
                if ta is None:
                    ta = a.get_block_dense(bi, bj)
                if tb is None:
                    tb = b.get_block_dense(bi, bj)
                ft = blockops.ewise_transpose(op, ta, tb)
                # routing receivers allocate out.dtype buffers
                locals_t[(bj, bi)] = ft.to(a.dtype)
        else:
            a = self._eval(ch)
            pm = a.presence()
            locals_t = {(bj, bi): blockops.transpose(t)
                        for (bi, bj), t in a.local_items()}

        out = BlockMatrix(a.ncols, a.nrows, a.block_size,
                          a.scheme.transpose(), a.dtype, a.device)
        rt = get_runtime()
        if not rt.is_distributed or a.scheme.is_replicated():
            out.blocks.update(locals_t)
            return out
        # Distributed: route each transposed block to the rank owning the
        # transposed index under out.scheme. pm is the (global) presence
        # of the UNtransposed source; the holder of block (bj, bi) of the
        # source is the one carrying transposed block (bi, bj) here.
        out_sch = out.scheme
        for (bi, bj) in out.block_indices():
            src = a.scheme.owner(bj, bi, rt.world_size, rt.grid)
            dst = out_sch.owner(bi, bj, rt.world_size, rt.grid)
            if not pm[bj, bi]:
                continue
            if src == dst:
                if rt.rank == src:
                    out.blocks[(bi, bj)] = locals_t[(bi, bj)]
            elif dst == -1:
                t = (locals_t[(bi, bj)] if rt.rank == src else
                     torch.empty(out.block_shape(bi, bj), dtype=out.dtype,
                                 device=out.device))
                dist.broadcast(t.contiguous(), src=src)
                out.blocks[(bi, bj)] = t
            else:
                if rt.rank == src:
                    dist.send(locals_t[(bi, bj)].contiguous(), dst=dst)
                elif rt.rank == dst:
                    t = torch.empty(out.block_shape(bi, bj), dtype=out.dtype,
                                    device=out.device)
                    dist.recv(t, src=src)
                    out.blocks[(bi, bj)] = t
        return out

    def _scalar(self, n: ScalarOp) -> BlockMatrix:
        a = self._eval(n.children[0])
        out = BlockMatrix(a.nrows, a.ncols, a.block_size, a.scheme,
                          a.dtype, a.device)
        op = {"add": blockops.SC_ADD, "mul": blockops.SC_MUL,
              "pow": blockops.SC_POW}[n.op]
        dense_fill = (n.op == "add" and n.s != 0.0)
        idxs = a.owned_indices() if dense_fill else \
            (idx for idx, _ in a.local_items())
        for idx in idxs:
            t = a.get_block_dense(*idx) if dense_fill else a.blocks[idx]
            out.blocks[idx] = blockops.ewise_scalar(op, t, n.s)
        return out

    def _ewise(self, n: Ewise) -> BlockMatrix:
        a = self._eval(n.children[0])
        b = self._eval(n.children[1])
        if a.block_size != b.block_size:
            raise ValueError("elementwise op needs equal block sizes")
        if a.scheme != b.scheme:
            # Follow the planner's move decision (physical.py Ewise rule:
            # align to the BIGGER operand's scheme so the smaller/sparser
            # side is the one that crosses xGMI).
            tgt = self.physical.get(n).scheme
            if a.scheme != tgt:
                a = a.repartition(tgt)
            if b.scheme != tgt:
                b = b.repartition(tgt)
        out = BlockMatrix(a.nrows, a.ncols, a.block_size, a.scheme,
                          a.dtype, a.device)
        op = {"add": blockops.EW_ADD, "sub": blockops.EW_SUB,
              "mul": blockops.EW_MUL, "div": blockops.EW_DIV,
              "min": blockops.EW_MIN, "max": blockops.EW_MAX}[n.op]
        if n.op in ("add", "sub", "min", "max"):
            keys = set(a.blocks) | set(b.blocks)
        elif n.op == "mul":
            keys = set(a.blocks) & set(b.blocks)
        else:  # div: need denominator everywhere it matters; compute on
            # union so zero/zero -> nan matches dense semantics on present
            # numerator blocks.
            keys = set(a.blocks)
        for idx in keys:
            ta = a.get_block(*idx)
            tb = b.get_block(*idx)
            if ta is None:
                ta = a.get_block_dense(*idx)
            if tb is None:
                tb = b.get_block_dense(*idx)
            out.blocks[idx] = blockops.ewise(op, ta, tb)
        return out

    def _matmul(self, n: MatMul) -> BlockMatrix:
        rt = get_runtime()
        pi = self.physical.get(n)
        a = self._eval(n.children[0])
        b = self._eval(n.children[1])
        if a.block_size != b.block_size:
            raise ValueError("matmul needs equal block sizes")
        algo = pi.mm_algo or "local"
        if not rt.is_distributed:
            out = BlockMatrix(a.nrows, b.ncols, a.block_size, a.scheme,
                              a.dtype, a.device)
            local_matmul(a, b, out)
            return out
        if algo == "summa":
            a = a.repartition(GRID) if a.scheme != GRID else a
            b = b.repartition(GRID) if b.scheme != GRID else b
            out = BlockMatrix(a.nrows, b.ncols, a.block_size, GRID,
                              a.dtype, a.device)
            summa_matmul(a, b, out)
            return out
        if algo == "bcast_b" or (algo == "local" and pi.scheme == ROW):
            if a.scheme != ROW:
                a = a.repartition(ROW)
            if not b.scheme.is_replicated():
                b = b.repartition(BCAST)
            out = BlockMatrix(a.nrows, b.ncols, a.block_size, ROW,
                              a.dtype, a.device)
            local_matmul(a, b, out)
            return out
        if algo == "bcast_a" or (algo == "local" and pi.scheme == COL):
            if not a.scheme.is_replicated():
                a = a.repartition(BCAST)
            if b.scheme != COL:
                b = b.repartition(COL)
            out = BlockMatrix(a.nrows, b.ncols, a.block_size, COL,
                              a.dtype, a.device)
            local_matmul(a, b, out)
            return out
        raise ValueError(f"unknown matmul algo {algo}")

    def _select(self, n: Select) -> BlockMatrix:
        a = self._eval(n.children[0])
        p = n.pred
        out = BlockMatrix(a.nrows, a.ncols, a.block_size, a.scheme,
                          a.dtype, a.device)
        bs = a.block_size
        rlo = p.row_lo if p.row_lo is not None else 0
        rhi = p.row_hi if p.row_hi is not None else a.nrows
        clo = p.col_lo if p.col_lo is not None else 0
        chi = p.col_hi if p.col_hi is not None else a.ncols
        for (bi, bj), t in a.local_items():
            r0, c0 = bi * bs, bj * bs
            r1, c1 = r0 + t.shape[0], c0 + t.shape[1]
            # block-granular pruning: fully outside -> drop (no kernel)
            if r1 <= rlo or r0 >= rhi or c1 <= clo or c0 >= chi:
                continue
            fully_inside = (r0 >= rlo and r1 <= rhi and
                            c0 >= clo and c1 <= chi)
            if fully_inside and not p.has_value():
                out.blocks[(bi, bj)] = t
                continue
            row_mask = col_mask = None
            if not fully_inside:
                rows = torch.arange(r0, r1)
                cols = torch.arange(c0, c1)
                row_mask = ((rows >= rlo) & (rows < rhi)).to(t.device)
                col_mask = ((cols >= clo) & (cols < chi)).to(t.device)
            res = blockops.mask_rows_cols(t, row_mask, col_mask) \
                if (row_mask is not None or col_mask is not None) else t
            if p.has_value():
                vlo = p.val_lo if p.val_lo is not None else -math.inf
                vhi = p.val_hi if p.val_hi is not None else math.inf
                res = blockops.mask_value(res, vlo, vhi, p.val_inside)
            out.blocks[(bi, bj)] = res
        return out

    def _agg(self, n: Agg) -> BlockMatrix:
        a = self._eval(n.children[0])
        rt = get_runtime()
        kind = n.kind
        dev = a.device
        red_kind = {"sum": blockops.RED_SUM, "avg": blockops.RED_SUM,
                    "max": blockops.RED_MAX, "min": blockops.RED_MIN,
                    "count": blockops.RED_NNZ, "nnz": blockops.RED_NNZ,
                    "trace": blockops.RED_SUM,
                    "norm_fro": blockops.RED_SUM}[kind]
        bs = a.block_size

        def items():
            # max/min must see absent blocks as zeros (they are logical
            # zeros); sum-like reductions can skip them.
            if kind in ("max", "min"):
                for idx in a.owned_indices():
                    yield idx, a.get_block_dense(*idx)
            else:
                yield from a.local_items()

        if n.axis is None:
            partial = torch.zeros(1, dtype=torch.float32, device=dev)
            init = None
            for (bi, bj), t in items():
                if kind == "trace":
                    if bi != bj:
                        continue
                    partial += blockops.trace_block(t)
                    continue
                if kind == "norm_fro":
                    partial += blockops.reduce2d(t, blockops.RED_SQSUM,
                                                 None)
                    continue
                v = blockops.reduce2d(t, red_kind, None)
                if kind in ("sum", "avg", "count", "nnz"):
                    partial += v
                else:  # max/min over blocks
                    init = v if init is None else (
                        torch.maximum(init, v) if kind == "max"
                        else torch.minimum(init, v))
            if kind in ("max", "min"):
                if init is None:
                    # This rank owns NO block indices at all (more grid
                    # rows/cols than block rows/cols under this scheme).
                    # The identity must be -/+inf: 0.0 would fabricate a
                    # candidate and win on an all-negative (max) /
                    # all-positive (min) matrix. Absent-but-owned blocks
                    # are already densified to zeros by items().
                    ident = -math.inf if kind == "max" else math.inf
                    init = torch.full((1,), ident, dtype=torch.float32,
                                      device=dev)
                partial = init.reshape(1)
            if rt.is_distributed and not a.scheme.is_replicated():
                op = {"max": dist.ReduceOp.MAX,
                      "min": dist.ReduceOp.MIN}.get(kind, dist.ReduceOp.SUM)
                dist.all_reduce(partial, op=op)
            if kind == "avg":
                partial = partial / (a.nrows * a.ncols)
            if kind == "norm_fro":
                partial = partial.sqrt()
            return _vector_result(partial.reshape(1, 1), a, (1, 1))

        # axis reductions -> vector results
        if n.axis == 1:  # per-row (m x 1)
            vec = _axis_identity(kind, a.nrows, dev)
            for (bi, bj), t in items():
                part = blockops.reduce2d(t, red_kind, 1)  # len h
                r0 = bi * bs
                _combine(vec, r0, part, kind)
        else:  # per-col (1 x n)
            vec = _axis_identity(kind, a.ncols, dev)
            for (bi, bj), t in items():
                part = blockops.reduce2d(t, red_kind, 0)
                c0 = bj * bs
                _combine(vec, c0, part, kind)
        if rt.is_distributed and not a.scheme.is_replicated():
            op = {"max": dist.ReduceOp.MAX,
                  "min": dist.ReduceOp.MIN}.get(kind, dist.ReduceOp.SUM)
            dist.all_reduce(vec, op=op)
        if kind == "avg":
            denom = a.ncols if n.axis == 1 else a.nrows
            vec = vec / denom
        shape = (a.nrows, 1) if n.axis == 1 else (1, a.ncols)
        return _vector_result(vec.reshape(shape), a, shape)

    def _gather_dense(self, n: Plan, what: str = "operand") -> torch.Tensor:
        from ..config import get_config
        limit = get_config().solve_max_side
        if max(n.shape) > limit:
            raise ValueError(
                f"{what} gathers the full {n.shape[0]}x{n.shape[1]} matrix "
                f"dense to every rank; the configured limit is "
                f"{limit}x{limit} (MATREL_SOLVE_MAX_SIDE). Solve/Inverse "
                f"are meant for small (e.g. Gram-matrix) operands.")
        m = self._eval(n)
        return m.to_dense()

    def _solve(self, n: Solve) -> BlockMatrix:
        from ..config import get_config
        sdt = (torch.float64 if get_config().solve_dtype == "fp64"
               else torch.float32)
        a_full = self._gather_dense(n.children[0], "solve").to(sdt)
        b_full = self._gather_dense(n.children[1], "solve rhs").to(sdt)
        # SPD fast path (X^T X of the OLS pipeline): Cholesky, fall back to
        # LU for general A (solve in fp32/fp64). cholesky
        # reads only the lower triangle, so gate it on exact symmetry -
        # a non-symmetric A with an SPD lower triangle would otherwise
        # silently solve the wrong system.
        if _is_symmetric(a_full):
            try:
                ch = torch.linalg.cholesky(a_full)
                x = torch.cholesky_solve(b_full, ch)
            except Exception:
                x = torch.linalg.solve(a_full, b_full)
        else:
            x = torch.linalg.solve(a_full, b_full)
        ref = self._eval(n.children[0])
        return _vector_result(x.float(), ref, tuple(x.shape))

    def _inverse(self, n: Inverse) -> BlockMatrix:
        from ..config import get_config
        sdt = (torch.float64 if get_config().solve_dtype == "fp64"
               else torch.float32)
        a_full = self._gather_dense(n.children[0], "inverse").to(sdt)
        # SPD fast path (X^T X of the OLS pipeline): Cholesky inverse -
        # rocprof showed LU (rocsolver getf2) at ~34% of config-5 time.
        # Symmetry-gated for the same reason as _solve.
        if _is_symmetric(a_full):
            try:
                ch = torch.linalg.cholesky(a_full)
                x = torch.cholesky_inverse(ch)
            except Exception:
                x = torch.linalg.inv(a_full)
        else:
            x = torch.linalg.inv(a_full)
        ref = self._eval(n.children[0])
        return _vector_result(x.float(), ref, tuple(x.shape))

    def _semijoin_value(self, n: SemiJoinValue) -> BlockMatrix:
        a = self._eval(n.children[0])
        b = self._eval(n.children[1])
        # gather B's distinct nonzero values (small by contract)
        bv = b.to_dense().reshape(-1)
        needles = torch.unique(bv[bv != 0].float())
        out = BlockMatrix(a.nrows, a.ncols, a.block_size, a.scheme,
                          a.dtype, a.device)
        for idx, t in a.local_items():
            out.blocks[idx] = blockops.mask_isin(t, needles)
        return out

    def _crossjoin(self, n: CrossJoin) -> BlockMatrix:
        a = self._eval(n.children[0])  # m x 1
        b = self._eval(n.children[1])  # 1 x n
        rt = get_runtime()
        # gather the two vectors (small relative to the m x n output)
        av = a.to_dense().reshape(-1)
        bv = b.to_dense().reshape(-1)
        out = BlockMatrix(a.nrows, b.ncols,
                          block_size=max(a.block_size, b.block_size),
                          scheme=GRID if rt.is_distributed else a.scheme,
                          device=a.device)
        bs = out.block_size
        for (bi, bj) in out.owned_indices():
            h, w = out.block_shape(bi, bj)
            asl = av[bi * bs: bi * bs + h]
            bsl = bv[bj * bs: bj * bs + w]
            t = blockops.outer(n.op, asl, bsl)
            out.blocks[(bi, bj)] = t.to(out.dtype)
        return out


# --------------------------------------------------------------------------

def _is_symmetric(a: torch.Tensor) -> bool:
    return (a.dim() == 2 and a.shape[0] == a.shape[1]
            and torch.equal(a, a.mT))


def _axis_identity(kind: str, length: int, dev) -> torch.Tensor:
    if kind == "max":
        return torch.full((length,), -math.inf, dtype=torch.float32,
                          device=dev)
    if kind == "min":
        return torch.full((length,), math.inf, dtype=torch.float32,
                          device=dev)
    return torch.zeros(length, dtype=torch.float32, device=dev)


def _combine(vec: torch.Tensor, off: int, part: torch.Tensor,
             kind: str) -> None:
    seg = vec[off: off + part.shape[0]]
    if kind == "max":
        torch.maximum(seg, part, out=seg)
    elif kind == "min":
        torch.minimum(seg, part, out=seg)
    else:
        seg += part


def _vector_result(t: torch.Tensor, like: BlockMatrix,
                   shape) -> BlockMatrix:
    """Wrap a small dense result (replicated on all ranks) as a
    BroadcastScheme BlockMatrix, tiled at the source matrix's block size so
    downstream ops (e.g. the OLS final multiply) see matching blocking."""
    out = BlockMatrix(shape[0], shape[1], block_size=like.block_size,
                      scheme=BCAST, dtype=torch.float32, device=like.device)
    t = t.float().to(like.device)
    bs = out.block_size
    for (bi, bj) in out.block_indices():
        h, w = out.block_shape(bi, bj)
        out.blocks[(bi, bj)] = t[bi * bs:bi * bs + h,
                                 bj * bs:bj * bs + w].contiguous()
    return out
