"""Physical planning: partition-scheme assignment + matmul algorithm choice.

Reference equivalent: the partition-scheme-selection pass and the physical
matmul strategies (SURVEY.md §2.1 "Optimizer: partition-scheme selection" and
"Physical matmul strategies": broadcast-MM | CPMM | RMM | partition-aware).

MI355X-native mapping:
  * broadcast-MM     -> replicate the small operand with ncclBroadcast,
                        other operand in row/col scheme, zero further comm.
  * CPMM / RMM       -> subsumed by SUMMA on the (Pr x Pc) xGMI grid:
                        per-k-step panel broadcasts on row/col
                        sub-communicators, local MFMA GEMM accumulate.
  * partition-aware  -> schemes already aligned => pure-local multiply.

The pass walks the optimized logical plan bottom-up, assigns every node an
output Scheme, and each MatMul an algorithm, minimizing estimated bytes
moved over xGMI (7 p2p links x ~153 GB/s per GPU: replication of a B-panel
to Pc-1 peers costs bytes*(Pc-1) link-seconds; SUMMA panels cost
N^2/Pr + N^2/Pc per step).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

from ..blocks.schemes import (BCAST, COL, GRID, ROW, Scheme)
from ..exec.runtime import get_runtime
from .logical import (Agg, CrossJoin, Ewise, Inverse, Leaf, MatMul, Plan,
                      ScalarOp, Select, SemiJoinValue, Solve, Transpose)

# Replication candidacy is a MEMORY guard only (the replicated operand must
# fit next to the shards in 288 GB HBM3E): configurable via
# MATREL_MAX_BCAST_BYTES. The bcast-vs-SUMMA decision itself is purely the
# estimated-xGMI-bytes comparison in _plan_matmul - broadcast-MM wins
# whenever rb < lb/pr + rb/pc (+ repartition moves), however large rb is.


@dataclass
class PhysInfo:
    scheme: Scheme
    mm_algo: Optional[str] = None      # 'local' | 'bcast_a' | 'bcast_b' | 'summa'
    est_flops: float = 0.0
    est_comm_bytes: float = 0.0


class PhysicalPlan:
    """Annotation map: logical node (by object identity) -> PhysInfo."""

    def __init__(self) -> None:
        self.info: Dict[int, PhysInfo] = {}

    def get(self, n: Plan) -> PhysInfo:
        return self.info[id(n)]

    def set(self, n: Plan, pi: PhysInfo) -> None:
        self.info[id(n)] = pi


def _nbytes(n: Plan, elem_bytes: int = 2) -> float:
    return n.est_nnz * elem_bytes


def plan_physical(root: Plan) -> PhysicalPlan:
    rt = get_runtime()
    pp = PhysicalPlan()
    single = not rt.is_distributed

    def visit(n: Plan) -> PhysInfo:
        if id(n) in pp.info:
            return pp.info[id(n)]
        ch = [visit(c) for c in n.children]

        if isinstance(n, Leaf):
            pi = PhysInfo(scheme=n.data.scheme)
        elif isinstance(n, Transpose):
            pi = PhysInfo(scheme=ch[0].scheme.transpose())
        elif isinstance(n, (ScalarOp, Select, SemiJoinValue)):
            pi = PhysInfo(scheme=ch[0].scheme)
        elif isinstance(n, Ewise):
            # align the smaller operand to the bigger one's scheme
            l, r = n.children
            if ch[0].scheme == ch[1].scheme:
                pi = PhysInfo(scheme=ch[0].scheme)
            else:
                move_right = _nbytes(r) <= _nbytes(l)
                tgt = ch[0].scheme if move_right else ch[1].scheme
                moved = _nbytes(r if move_right else l)
                pi = PhysInfo(scheme=tgt, est_comm_bytes=moved)
        elif isinstance(n, CrossJoin):
            pi = PhysInfo(scheme=GRID if not single else ch[0].scheme)
        elif isinstance(n, MatMul):
            pi = _plan_matmul(n, ch, single)
        elif isinstance(n, Agg):
            # aggregates reduce to a small replicated result
            pi = PhysInfo(scheme=BCAST,
                          est_comm_bytes=0 if single else n.est_nnz * 4)
        elif isinstance(n, (Solve, Inverse)):
            pi = PhysInfo(scheme=BCAST,
                          est_comm_bytes=0 if single else
                          _nbytes(n.children[0], 4))
        else:
            raise TypeError(f"no physical rule for {type(n).__name__}")
        pp.set(n, pi)
        return pi

    def _plan_matmul(n: MatMul, ch, single: bool) -> PhysInfo:
        l, r = n.children
        if single:
            return PhysInfo(scheme=ch[0].scheme, mm_algo="local",
                            est_flops=2.0 * l.nrows * l.ncols * r.ncols)
        flops = 2.0 * l.nrows * l.ncols * r.ncols \
            * max(l.density, 1e-6) * max(r.density, 1e-6)
        lb, rb = _nbytes(l), _nbytes(r)
        rt = get_runtime()
        pr, pc = rt.grid
        w = rt.world_size
        # candidate costs in bytes moved per GPU (xGMI is p2p; a broadcast
        # of x bytes to g-1 peers is ~x per link with tree/multicast)
        cands = []
        # partition-aware local: A row-partitioned, B replicated already
        if ch[0].scheme == ROW and ch[1].scheme == BCAST:
            cands.append(("local", 0.0, ROW))
        if ch[0].scheme == BCAST and ch[1].scheme == COL:
            cands.append(("local", 0.0, COL))
        from ..config import get_config
        mem_cap = get_config().max_bcast_bytes
        if rb <= mem_cap:
            move_a = 0.0 if ch[0].scheme == ROW else lb / w
            cands.append(("bcast_b", rb + move_a, ROW))
        if lb <= mem_cap:
            move_b = 0.0 if ch[1].scheme == COL else rb / w
            cands.append(("bcast_a", lb + move_b, COL))
        # SUMMA: every GPU receives A-panels (N*K/pr of A rows it shares)
        # and B-panels across the k loop: bytes ~ lb/pr + rb/pc
        move_a = 0.0 if ch[0].scheme == GRID else lb / w
        move_b = 0.0 if ch[1].scheme == GRID else rb / w
        cands.append(("summa", lb / pr + rb / pc + move_a + move_b, GRID))
        algo, bytes_, out_scheme = min(cands, key=lambda c: c[1])
        return PhysInfo(scheme=out_scheme, mm_algo=algo, est_flops=flops,
                        est_comm_bytes=bytes_)

    visit(root)
    return pp
