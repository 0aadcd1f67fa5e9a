"""Matrix-expression DSL: user-facing lazy Matrix handles.

Reference equivalent: MatRel's Dataset-like matrix wrapper with one method
per DSL verb (SURVEY.md §2.1 "Matrix-expression DSL": multiply, transpose,
scalar +/*, element-wise +/*//, pow; relational verbs: select by
index/value, aggregate sum/trace/rowSum/colSum/count/avg/max/min,
cross-product join).

A Matrix wraps a logical Plan node. Operators build the tree lazily;
`.compute()` runs the optimizer batch + physical planner + executor and
returns a materialized Matrix (Leaf). Scalars come back via `.item()`,
small tensors via `.to_tensor()`.
"""

from __future__ import annotations

from typing import Optional, Tuple, Union

import torch

from .blocks.matrix import BlockMatrix
from .plan import logical as L
from .plan.optimizer import optimize

Number = Union[int, float]


class Matrix:
    def __init__(self, plan: L.Plan) -> None:
        self.plan = plan

    # --- construction shortcuts ----------------------------------------

    @staticmethod
    def from_blocks(bm: BlockMatrix, name: Optional[str] = None,
                    density: Optional[float] = None) -> "Matrix":
        return Matrix(L.Leaf(bm, name=name, density=density))

    # --- shape ----------------------------------------------------------

    @property
    def shape(self) -> Tuple[int, int]:
        return self.plan.shape

    @property
    def nrows(self) -> int:
        return self.plan.nrows

    @property
    def ncols(self) -> int:
        return self.plan.ncols

    # --- linear algebra verbs -------------------------------------------

    def multiply(self, other: "Matrix") -> "Matrix":
        return Matrix(L.MatMul(self.plan, other.plan))

    def __matmul__(self, other: "Matrix") -> "Matrix":
        return self.multiply(other)

    def transpose(self) -> "Matrix":
        return Matrix(L.Transpose(self.plan))

    @property
    def T(self) -> "Matrix":
        return self.transpose()

    def t(self) -> "Matrix":
        return self.transpose()

    def matpow(self, k: int) -> "Matrix":
        return Matrix(L.MatPower(self.plan, k))

    def elem_pow(self, s: Number) -> "Matrix":
        return Matrix(L.ScalarOp(self.plan, "pow", float(s)))

    def __pow__(self, s: Number) -> "Matrix":
        # Reference MatrixPowerOperator semantics: ELEMENTWISE power.
        return self.elem_pow(s)

    # --- scalar & elementwise -------------------------------------------

    def _binary(self, other, op: str) -> "Matrix":
        if isinstance(other, (int, float)):
            if op == "add":
                return Matrix(L.ScalarOp(self.plan, "add", float(other)))
            if op == "sub":
                return Matrix(L.ScalarOp(self.plan, "add", -float(other)))
            if op == "mul":
                return Matrix(L.ScalarOp(self.plan, "mul", float(other)))
            if op == "div":
                return Matrix(L.ScalarOp(self.plan, "mul", 1.0 / other))
        return Matrix(L.Ewise(self.plan, other.plan, op))

    def __add__(self, o):
        return self._binary(o, "add")

    def __radd__(self, o):
        return self._binary(o, "add")

    def __sub__(self, o):
        return self._binary(o, "sub")

    def __rsub__(self, o):
        return Matrix(L.ScalarOp(
            L.ScalarOp(self.plan, "mul", -1.0), "add", float(o)))

    def __mul__(self, o):
        return self._binary(o, "mul")

    def __rmul__(self, o):
        return self._binary(o, "mul")

    def __truediv__(self, o):
        return self._binary(o, "div")

    def __rtruediv__(self, o):
        # s / A == s * A**(-1) elementwise
        return Matrix(L.ScalarOp(
            L.ScalarOp(self.plan, "pow", -1.0), "mul", float(o)))

    def __neg__(self):
        return Matrix(L.ScalarOp(self.plan, "mul", -1.0))

    # --- relational verbs -----------------------------------------------

    def select(self, rows: Optional[Tuple[Optional[int], Optional[int]]] = None,
               cols: Optional[Tuple[Optional[int], Optional[int]]] = None,
               values: Optional[Tuple[Optional[float], Optional[float]]] = None,
               inside: bool = True) -> "Matrix":
        """Keep entries whose row index is in [rows), col index in [cols)
        and value in [values] (or outside when inside=False); everything
        else becomes zero. Whole excluded blocks are pruned at plan time."""
        pred = L.SelectPredicate(
            row_lo=None if rows is None else rows[0],
            row_hi=None if rows is None else rows[1],
            col_lo=None if cols is None else cols[0],
            col_hi=None if cols is None else cols[1],
            val_lo=None if values is None else values[0],
            val_hi=None if values is None else values[1],
            val_inside=inside)
        return Matrix(L.Select(self.plan, pred))

    def select_rows(self, lo: int, hi: int) -> "Matrix":
        return self.select(rows=(lo, hi))

    def select_cols(self, lo: int, hi: int) -> "Matrix":
        return self.select(cols=(lo, hi))

    def select_values(self, lo: float, hi: float,
                      inside: bool = True) -> "Matrix":
        return self.select(values=(lo, hi), inside=inside)

    def cross_join(self, other: "Matrix", op: str = "mul") -> "Matrix":
        return Matrix(L.CrossJoin(self.plan, other.plan, op))

    def join_elem(self, other: "Matrix", op: str = "mul") -> "Matrix":
        """Index-equality join on (i,j): combine co-located entries with
        op in {mul, add, min, max} (MatRel join on indices)."""
        return Matrix(L.Ewise(self.plan, other.plan, op))

    def join_rows(self, other: "Matrix") -> "Matrix":
        """Join on row index with multiply-aggregate combiner:
        C[j,k] = sum_i A[i,j]*B[i,k] == A^T @ B (lowered to the MFMA
        matmul path; chain/scheme optimizations apply)."""
        return Matrix(L.MatMul(L.Transpose(self.plan), other.plan))

    def join_cols(self, other: "Matrix") -> "Matrix":
        """Join on column index: C[i,k] = sum_j A[i,j]*B[k,j] == A @ B^T."""
        return Matrix(L.MatMul(self.plan, L.Transpose(other.plan)))

    def semi_join_values(self, other: "Matrix") -> "Matrix":
        """Keep entries of self whose value occurs among other's nonzero
        values (MatRel join on values, semi-join form)."""
        return Matrix(L.SemiJoinValue(self.plan, other.plan))

    def elem_min(self, other: "Matrix") -> "Matrix":
        return Matrix(L.Ewise(self.plan, other.plan, "min"))

    def elem_max(self, other: "Matrix") -> "Matrix":
        return Matrix(L.Ewise(self.plan, other.plan, "max"))

    # --- aggregates ------------------------------------------------------

    def _agg(self, kind: str, axis: Optional[int]) -> "Matrix":
        return Matrix(L.Agg(self.plan, kind, axis))

    def sum(self) -> "Matrix":
        return self._agg("sum", None)

    def avg(self) -> "Matrix":
        return self._agg("avg", None)

    def max(self) -> "Matrix":
        return self._agg("max", None)

    def min(self) -> "Matrix":
        return self._agg("min", None)

    def count(self) -> "Matrix":
        """Number of nonzero entries (reference 'count')."""
        return self._agg("count", None)

    def nnz(self) -> "Matrix":
        return self._agg("nnz", None)

    def trace(self) -> "Matrix":
        return self._agg("trace", None)

    def norm_fro(self) -> "Matrix":
        return self._agg("norm_fro", None)

    def row_norms(self) -> "Matrix":
        """Per-row L2 norms (m x 1): sqrt(rowsums(A*A))."""
        return ((self * self).row_sums()) ** 0.5

    def col_norms(self) -> "Matrix":
        """Per-column L2 norms (1 x n)."""
        return ((self * self).col_sums()) ** 0.5

    def row_sums(self) -> "Matrix":
        return self._agg("sum", 1)

    def col_sums(self) -> "Matrix":
        return self._agg("sum", 0)

    def row_maxs(self) -> "Matrix":
        return self._agg("max", 1)

    def col_maxs(self) -> "Matrix":
        return self._agg("max", 0)

    def row_mins(self) -> "Matrix":
        return self._agg("min", 1)

    def col_mins(self) -> "Matrix":
        return self._agg("min", 0)

    def row_avgs(self) -> "Matrix":
        return self._agg("avg", 1)

    def col_avgs(self) -> "Matrix":
        return self._agg("avg", 0)

    def row_counts(self) -> "Matrix":
        return self._agg("count", 1)

    def col_counts(self) -> "Matrix":
        return self._agg("count", 0)

    # --- solves ----------------------------------------------------------

    def solve(self, b: "Matrix") -> "Matrix":
        """X with self @ X = b (small square self)."""
        return Matrix(L.Solve(self.plan, b.plan))

    def inverse(self) -> "Matrix":
        return Matrix(L.Inverse(self.plan))

    # --- execution --------------------------------------------------------

    def optimized_plan(self) -> L.Plan:
        return optimize(self.plan)

    def compile(self):
        """Capture this expression's kernel sequence in a hipGraph and
        return a replayable callable (exec/graphcap.py). Falls back to
        eager when the plan is not capturable."""
        from .exec.graphcap import compile_expression
        return compile_expression(self)

    def compute(self) -> "Matrix":
        """Optimize + plan + execute; returns a materialized Matrix."""
        from .exec.executor import Executor
        opt = optimize(self.plan)
        result = Executor().execute(opt)
        return Matrix(L.Leaf(result))

    def explain(self, verbose: bool = True) -> str:
        """Plan inspection (the reference's Spark `explain()` analog):
        the logical tree, the optimized tree, and per-node physical
        annotations (partition scheme, matmul algorithm, estimated flops
        and xGMI bytes). Prints when verbose; always returns the text."""
        from .plan.physical import plan_physical
        opt = optimize(self.plan)
        pp = plan_physical(opt)
        lines = ["== logical plan ==", self.plan.pretty().rstrip(),
                 "== optimized plan ==", opt.pretty().rstrip(),
                 "== physical annotations =="]

        seen = set()

        def walk(n: L.Plan, indent: int) -> None:
            pi = pp.get(n)
            extra = f" algo={pi.mm_algo}" if pi.mm_algo else ""
            stats = ""
            if pi.est_flops:
                stats += f" ~{pi.est_flops / 1e9:.2f} GF"
            if pi.est_comm_bytes:
                stats += f" ~{pi.est_comm_bytes / 1e6:.2f} MB moved"
            if id(n) in seen and n.children:
                # CSE-shared subtree: executes once, memoized
                lines.append("  " * indent + f"{type(n).__name__} "
                             f"{n.shape} (cse-shared, see above)")
                return
            seen.add(id(n))
            lines.append("  " * indent + f"{type(n).__name__} "
                         f"{n.shape} scheme={pi.scheme}{extra}{stats}")
            for c in n.children:
                walk(c, indent + 1)

        walk(opt, 0)
        text = "\n".join(lines)
        if verbose:
            print(text)
        return text

    # --- materialized accessors ----------------------------------------

    def _materialized(self) -> BlockMatrix:
        if not isinstance(self.plan, L.Leaf):
            return self.compute()._materialized()
        return self.plan.data

    @property
    def blocks(self) -> BlockMatrix:
        return self._materialized()

    def to_tensor(self) -> torch.Tensor:
        """Full dense tensor (collective; small matrices / tests)."""
        return self._materialized().to_dense()

    def item(self) -> float:
        t = self.to_tensor()
        assert t.numel() == 1, f"item() on shape {tuple(t.shape)}"
        return float(t.reshape(()).item())

    def save(self, path: str) -> None:
        from .io.format import save_matrix
        save_matrix(self._materialized(), path)

    def save_coo(self, path: str) -> None:
        from .io.format import save_coo_text
        save_coo_text(self._materialized(), path)

    def __repr__(self) -> str:
        return f"Matrix({self.nrows}x{self.ncols}, plan={type(self.plan).__name__})"
