"""Logical plan IR for matrix expressions.

Reference equivalent: MatRel's Catalyst TreeNode operators - one logical
class per DSL verb (SURVEY.md §2.1 "Logical operators": TransposeOperator,
MatrixScalarAddOperator, MatrixMatrixMultiplicationOperator, SelectOperator,
AggregateOperator, ...). Here the IR is a small immutable-ish tree of Python
dataclass-like nodes with:

  * shape inference,
  * sparsity (density) estimation feeding the cost-based optimizer
    (SURVEY.md §2.1 "Optimizer: matrix-chain ordering" - nnz propagation),
  * structural hashing for common-subexpression elimination.

The tree is built by the DSL (matrel_amd/dsl.py), rewritten by
plan/optimizer.py, and lowered/executed by exec/executor.py.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

from ..blocks.matrix import BlockMatrix


class Plan:
    """Base logical operator."""

    children: List["Plan"]

    def __init__(self, children: List["Plan"]) -> None:
        self.children = children

    # -- shape ----------------------------------------------------------
    @property
    def shape(self) -> Tuple[int, int]:
        raise NotImplementedError

    @property
    def nrows(self) -> int:
        return self.shape[0]

    @property
    def ncols(self) -> int:
        return self.shape[1]

    # -- sparsity -------------------------------------------------------
    @property
    def density(self) -> float:
        """Estimated fraction of nonzero entries (nnz / (m*n))."""
        raise NotImplementedError

    @property
    def est_nnz(self) -> float:
        return self.density * self.nrows * self.ncols

    # -- structural identity (CSE) --------------------------------------
    def key(self) -> tuple:
        raise NotImplementedError

    def __hash__(self) -> int:
        return hash(self.key())

    def __eq__(self, other: object) -> bool:
        return isinstance(other, Plan) and self.key() == other.key()

    def with_children(self, children: List["Plan"]) -> "Plan":
        raise NotImplementedError

    def pretty(self, indent: int = 0) -> str:
        name = type(self).__name__
        extra = getattr(self, "_pretty_extra", lambda: "")()
        s = "  " * indent + f"{name}{extra} {self.shape}\n"
        for c in self.children:
            s += c.pretty(indent + 1)
        return s


class Leaf(Plan):
    """A materialized BlockMatrix operand."""

    _ids = 0

    def __init__(self, data: BlockMatrix, name: Optional[str] = None,
                 density: Optional[float] = None) -> None:
        super().__init__([])
        self.data = data
        if name is None:
            Leaf._ids += 1
            name = f"m{Leaf._ids}"
        self.name = name
        # Density: explicit hint, else block-level presence fraction.
        # (Within-block density is refined by io/format metadata when the
        # matrix was loaded from disk.)
        self._density = density

    @property
    def shape(self) -> Tuple[int, int]:
        return self.data.shape

    @property
    def density(self) -> float:
        if self._density is not None:
            return self._density
        # GLOBAL block-presence fraction (one cached all_reduce), NOT the
        # local block count: the planner runs replicated on every rank and
        # every cost-model input must be rank-uniform, or ranks pick
        # different matmul algorithms and the collectives deadlock (seen
        # at world 4 with ragged shapes: summa vs bcast_b cost inversion).
        total = self.data.grid_rows * self.data.grid_cols
        if total == 0:
            return 1.0
        d = self.data.block_density()
        self._density = max(d, 1.0 / total)
        return self._density

    def key(self) -> tuple:
        return ("leaf", id(self.data))

    def with_children(self, children: List[Plan]) -> Plan:
        return self

    def _pretty_extra(self) -> str:
        return f"[{self.name}]"


class Transpose(Plan):
    def __init__(self, c: Plan) -> None:
        super().__init__([c])

    @property
    def shape(self) -> Tuple[int, int]:
        r, c = self.children[0].shape
        return (c, r)

    @property
    def density(self) -> float:
        return self.children[0].density

    def key(self) -> tuple:
        return ("t", self.children[0].key())

    def with_children(self, ch: List[Plan]) -> Plan:
        return Transpose(ch[0])


class ScalarOp(Plan):
    """A + s, A * s, A ** s (elementwise power by a scalar)."""

    def __init__(self, c: Plan, op: str, s: float) -> None:
        assert op in ("add", "mul", "pow")
        super().__init__([c])
        self.op = op
        self.s = float(s)

    @property
    def shape(self) -> Tuple[int, int]:
        return self.children[0].shape

    @property
    def density(self) -> float:
        d = self.children[0].density
        if self.op == "add" and self.s != 0.0:
            return 1.0
        if self.op == "mul" and self.s == 0.0:
            return 0.0
        return d

    def key(self) -> tuple:
        return ("s" + self.op, self.s, self.children[0].key())

    def with_children(self, ch: List[Plan]) -> Plan:
        return ScalarOp(ch[0], self.op, self.s)

    def _pretty_extra(self) -> str:
        return f"[{self.op} {self.s}]"


class Ewise(Plan):
    """Elementwise A op B; shapes must match. Ops min/max realize the
    reference's index-equality join with min/max combiners (an index join
    on (i,j)=(i,j) IS an elementwise combine - SURVEY.md §2.1 joins)."""

    def __init__(self, l: Plan, r: Plan, op: str) -> None:
        assert op in ("add", "sub", "mul", "div", "min", "max")
        if l.shape != r.shape:
            raise ValueError(f"ewise {op}: shape {l.shape} vs {r.shape}")
        super().__init__([l, r])
        self.op = op

    @property
    def shape(self) -> Tuple[int, int]:
        return self.children[0].shape

    @property
    def density(self) -> float:
        da, db = self.children[0].density, self.children[1].density
        if self.op in ("add", "sub", "min", "max"):
            return min(1.0, da + db - da * db)
        if self.op == "mul":
            return da * db
        return da  # div: zeros of numerator survive

    def key(self) -> tuple:
        k = (self.children[0].key(), self.children[1].key())
        if self.op in ("add", "mul", "min", "max"):  # commutative
            # keys can contain non-orderable payloads (SelectPredicate):
            # canonicalize by repr, which is deterministic for these tuples
            k = tuple(sorted(k, key=repr))
        return ("e" + self.op,) + k

    def with_children(self, ch: List[Plan]) -> Plan:
        return Ewise(ch[0], ch[1], self.op)

    def _pretty_extra(self) -> str:
        return f"[{self.op}]"


class MatMul(Plan):
    def __init__(self, l: Plan, r: Plan) -> None:
        if l.ncols != r.nrows:
            raise ValueError(f"matmul: {l.shape} @ {r.shape}")
        super().__init__([l, r])

    @property
    def shape(self) -> Tuple[int, int]:
        return (self.children[0].nrows, self.children[1].ncols)

    @property
    def density(self) -> float:
        # nnz propagation for C = A @ B with inner dim k:
        # P(c_ij != 0) ~= 1 - (1 - da*db)^k   (independent-nonzeros model,
        # the standard estimator the reference's sparsity-aware cost model
        # uses per SURVEY.md §2.1)
        da, db = self.children[0].density, self.children[1].density
        k = self.children[0].ncols
        p = da * db
        if p <= 0:
            return 0.0
        if p * k > 20:
            return 1.0
        return 1.0 - (1.0 - p) ** k

    def key(self) -> tuple:
        return ("mm", self.children[0].key(), self.children[1].key())

    def with_children(self, ch: List[Plan]) -> Plan:
        return MatMul(ch[0], ch[1])


class MatPower(Plan):
    """A @ A @ ... @ A (k times), A square. Lowered by the optimizer."""

    def __init__(self, c: Plan, k: int) -> None:
        if c.nrows != c.ncols:
            raise ValueError("matrix power needs a square matrix")
        assert k >= 1
        super().__init__([c])
        self.k = int(k)

    @property
    def shape(self) -> Tuple[int, int]:
        return self.children[0].shape

    @property
    def density(self) -> float:
        d = self.children[0].density
        n = self.nrows
        for _ in range(self.k - 1):
            p = d * self.children[0].density
            d = 1.0 if p * n > 20 else 1.0 - (1.0 - p) ** n
        return d

    def key(self) -> tuple:
        return ("pow", self.k, self.children[0].key())

    def with_children(self, ch: List[Plan]) -> Plan:
        return MatPower(ch[0], self.k)

    def _pretty_extra(self) -> str:
        return f"[k={self.k}]"


@dataclass(frozen=True)
class SelectPredicate:
    """Conjunction of interval predicates on row index, col index, value.

    None bound = unbounded. Index bounds are half-open [lo, hi).
    val_inside=False selects values OUTSIDE [val_lo, val_hi].
    Reference: MatRel SelectOperator (selection on matrix indices/values,
    SURVEY.md §2.1 / §3.4).
    """
    row_lo: Optional[int] = None
    row_hi: Optional[int] = None
    col_lo: Optional[int] = None
    col_hi: Optional[int] = None
    val_lo: Optional[float] = None
    val_hi: Optional[float] = None
    val_inside: bool = True

    def has_index(self) -> bool:
        return any(v is not None for v in
                   (self.row_lo, self.row_hi, self.col_lo, self.col_hi))

    def has_value(self) -> bool:
        return self.val_lo is not None or self.val_hi is not None


class Select(Plan):
    """Selection: entries failing the predicate become zero (the reference
    keeps matrix shape and zeroes non-selected entries; whole failing
    blocks are pruned at plan/exec time)."""

    def __init__(self, c: Plan, pred: SelectPredicate) -> None:
        super().__init__([c])
        self.pred = pred

    @property
    def shape(self) -> Tuple[int, int]:
        return self.children[0].shape

    @property
    def density(self) -> float:
        d = self.children[0].density
        m, n = self.shape
        p = self.pred
        if p.has_index():
            rl, rh = p.row_lo or 0, p.row_hi if p.row_hi is not None else m
            cl, ch = p.col_lo or 0, p.col_hi if p.col_hi is not None else n
            frac = max(0, rh - rl) / m * max(0, ch - cl) / n
            d *= frac
        if p.has_value():
            d *= 0.5  # no value histogram: flat 50% selectivity guess
        return d

    def key(self) -> tuple:
        return ("sel", self.pred, self.children[0].key())

    def with_children(self, ch: List[Plan]) -> Plan:
        return Select(ch[0], self.pred)

    def _pretty_extra(self) -> str:
        return f"[{self.pred}]"


class Agg(Plan):
    """Aggregation: kind in {sum,avg,max,min,count,trace,nnz,norm_fro};
    axis None=whole matrix (1x1 result), 0=per-column (1xn), 1=per-row (mx1).
    Reference: MatRel AggregateOperator (rowSum/colSum/trace/count/...).
    """

    KINDS = ("sum", "avg", "max", "min", "count", "trace", "nnz", "norm_fro")

    def __init__(self, c: Plan, kind: str, axis: Optional[int]) -> None:
        assert kind in self.KINDS
        if kind == "trace" and c.nrows != c.ncols:
            raise ValueError("trace needs a square matrix")
        super().__init__([c])
        self.kind = kind
        self.axis = axis

    @property
    def shape(self) -> Tuple[int, int]:
        if self.axis is None or self.kind == "trace":
            return (1, 1)
        if self.axis == 0:
            return (1, self.children[0].ncols)
        return (self.children[0].nrows, 1)

    @property
    def density(self) -> float:
        return 1.0

    def key(self) -> tuple:
        return ("agg", self.kind, self.axis, self.children[0].key())

    def with_children(self, ch: List[Plan]) -> Plan:
        return Agg(ch[0], self.kind, self.axis)

    def _pretty_extra(self) -> str:
        return f"[{self.kind} axis={self.axis}]"


class Solve(Plan):
    """X = A^-1 B for a SMALL square A (gathered, dense solve in fp32/fp64).

    Reference: the OLS pipeline's small (k x k) inverse computed at the
    driver. A must fit on one device.
    """

    def __init__(self, a: Plan, b: Plan) -> None:
        if a.nrows != a.ncols or a.ncols != b.nrows:
            raise ValueError(f"solve: {a.shape} \\ {b.shape}")
        super().__init__([a, b])

    @property
    def shape(self) -> Tuple[int, int]:
        return (self.children[0].ncols, self.children[1].ncols)

    @property
    def density(self) -> float:
        return 1.0

    def key(self) -> tuple:
        return ("solve", self.children[0].key(), self.children[1].key())

    def with_children(self, ch: List[Plan]) -> Plan:
        return Solve(ch[0], ch[1])


class Inverse(Plan):
    """A^-1 for a SMALL square A (gathered dense inverse)."""

    def __init__(self, a: Plan) -> None:
        if a.nrows != a.ncols:
            raise ValueError("inverse needs square")
        super().__init__([a])

    @property
    def shape(self) -> Tuple[int, int]:
        return self.children[0].shape

    @property
    def density(self) -> float:
        return 1.0

    def key(self) -> tuple:
        return ("inv", self.children[0].key())

    def with_children(self, ch: List[Plan]) -> Plan:
        return Inverse(ch[0])


class CrossJoin(Plan):
    """Cross-product join on values with a combiner: every entry of A
    against every entry of B is conceptually paired; MatRel restricts to
    the rank-structured case realized as outer products. We expose the
    common case used by the reference workloads: join A (m x 1) with
    B (1 x n) producing C[i,j] = f(A[i], B[j]) for f in {mul, add}.
    (General index-equality joins are the join_elem/join_rows/join_cols
    verbs, executed block-wise in the executor.)
    """

    def __init__(self, l: Plan, r: Plan, op: str = "mul") -> None:
        if l.ncols != 1 or r.nrows != 1:
            raise ValueError("cross join expects (m x 1) x (1 x n)")
        assert op in ("mul", "add")
        super().__init__([l, r])
        self.op = op

    @property
    def shape(self) -> Tuple[int, int]:
        return (self.children[0].nrows, self.children[1].ncols)

    @property
    def density(self) -> float:
        da, db = self.children[0].density, self.children[1].density
        return da * db if self.op == "mul" else min(1.0, da + db)

    def key(self) -> tuple:
        return ("xjoin", self.op, self.children[0].key(),
                self.children[1].key())

    def with_children(self, ch: List[Plan]) -> Plan:
        return CrossJoin(ch[0], ch[1], self.op)


class SemiJoinValue(Plan):
    """Value join (semi-join form): keep entries of A whose VALUE appears
    among the nonzero values of B; zero the rest. Reference: MatRel's join
    on matrix values. B must be small enough to gather
    its distinct values host-side (the common analytics case)."""

    def __init__(self, l: Plan, r: Plan) -> None:
        super().__init__([l, r])

    @property
    def shape(self):
        return self.children[0].shape

    @property
    def density(self) -> float:
        return self.children[0].density * 0.5

    def key(self) -> tuple:
        return ("semijoinv", self.children[0].key(), self.children[1].key())

    def with_children(self, ch):
        return SemiJoinValue(ch[0], ch[1])
