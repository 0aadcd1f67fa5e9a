"""Rule-based + cost-based optimizer over the logical plan.

Reference equivalent: MatfastOptimizer's Catalyst rule batches
(SURVEY.md §2.1 "Optimizer" rows):
  * transpose push-down        ((AB)^T -> B^T A^T, (A^T)^T -> A, ...)
  * scalar / elementwise folding
  * selection push-down to block granularity
  * matrix-chain multiplication reordering (DP, sparsity-aware cost)
  * common-subexpression elimination (hash-consing)

Each pass is a pure function Plan -> Plan; `optimize` runs the batch in a
fixed order (cheap rewrites first so the DP sees the normalized tree).
Passes can be toggled via MatrelConfig for tests/ablation.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

from ..config import get_config
from .logical import (Agg, Ewise, MatMul, MatPower, Plan, ScalarOp,
                      Select, SelectPredicate, Transpose)


# --------------------------------------------------------------------------
# generic bottom-up rewriter
# --------------------------------------------------------------------------

def _rewrite_bottom_up(p: Plan, rule) -> Plan:
    new_children = [_rewrite_bottom_up(c, rule) for c in p.children]
    if any(a is not b for a, b in zip(new_children, p.children)):
        p = p.with_children(new_children)
    return rule(p)


# --------------------------------------------------------------------------
# pass: matrix power lowering (A^k via exponentiation by squaring)
# --------------------------------------------------------------------------

def lower_matpower(p: Plan) -> Plan:
    def rule(n: Plan) -> Plan:
        if not isinstance(n, MatPower):
            return n
        base, k = n.children[0], n.k
        # Exponentiation by squaring; shared squares are CSE'd later.
        result = None
        sq = base
        while k > 0:
            if k & 1:
                result = sq if result is None else MatMul(result, sq)
            k >>= 1
            if k:
                sq = MatMul(sq, sq)
        return result
    return _rewrite_bottom_up(p, rule)


# --------------------------------------------------------------------------
# pass: transpose push-down
# --------------------------------------------------------------------------

def push_transpose(p: Plan) -> Plan:
    """Normalize so Transpose appears only directly above leaves.

    (A^T)^T -> A;  (AB)^T -> B^T A^T;  (A op B)^T -> A^T op B^T;
    (s*A)^T -> s*(A^T);  select/agg commute with transpose by swapping
    their row/col roles.
    """
    def rule(n: Plan) -> Plan:
        if not isinstance(n, Transpose):
            return n
        c = n.children[0]
        if isinstance(c, Transpose):
            return c.children[0]
        if isinstance(c, MatMul):
            return rule_wrap(MatMul(push_t(c.children[1]),
                                    push_t(c.children[0])))
        if isinstance(c, Ewise):
            return rule_wrap(Ewise(push_t(c.children[0]),
                                   push_t(c.children[1]), c.op))
        if isinstance(c, ScalarOp):
            return rule_wrap(ScalarOp(push_t(c.children[0]), c.op, c.s))
        if isinstance(c, Select):
            pr = c.pred
            swapped = SelectPredicate(
                row_lo=pr.col_lo, row_hi=pr.col_hi,
                col_lo=pr.row_lo, col_hi=pr.row_hi,
                val_lo=pr.val_lo, val_hi=pr.val_hi,
                val_inside=pr.val_inside)
            return rule_wrap(Select(push_t(c.children[0]), swapped))
        return n

    def push_t(x: Plan) -> Plan:
        return rule(Transpose(x))

    def rule_wrap(x: Plan) -> Plan:
        return _rewrite_bottom_up(x, rule)

    return _rewrite_bottom_up(p, rule)


# --------------------------------------------------------------------------
# pass: scalar folding
# --------------------------------------------------------------------------

def fold_scalars(p: Plan) -> Plan:
    def rule(n: Plan) -> Plan:
        if not isinstance(n, ScalarOp):
            return n
        c = n.children[0]
        # identity ops
        if n.op == "mul" and n.s == 1.0:
            return c
        if n.op == "add" and n.s == 0.0:
            return c
        if n.op == "pow" and n.s == 1.0:
            return c
        if isinstance(c, ScalarOp):
            if n.op == c.op == "mul":
                return ScalarOp(c.children[0], "mul", n.s * c.s)
            if n.op == c.op == "add":
                return ScalarOp(c.children[0], "add", n.s + c.s)
            if n.op == c.op == "pow":
                return ScalarOp(c.children[0], "pow", n.s * c.s)
        # pull scalar-mul out of matmul operands: (aA)(bB) handled via
        # MatMul rule below would need product collection; the common single
        # case (s*A) @ B -> s*(A @ B) is done in the MatMul rule.
        return n

    def mm_rule(n: Plan) -> Plan:
        if isinstance(n, MatMul):
            l, r = n.children
            s = 1.0
            if isinstance(l, ScalarOp) and l.op == "mul":
                s *= l.s
                l = l.children[0]
            if isinstance(r, ScalarOp) and r.op == "mul":
                s *= r.s
                r = r.children[0]
            if s != 1.0:
                return ScalarOp(MatMul(l, r), "mul", s)
        return rule(n)

    return _rewrite_bottom_up(p, mm_rule)


# --------------------------------------------------------------------------
# pass: selection push-down
# --------------------------------------------------------------------------

def push_select(p: Plan) -> Plan:
    """Push index-range selections toward leaves so block pruning happens
    as early as possible (prune whole blocks at plan time).

    Select(A @ B, rows r, cols c) -> Select(A, rows r) @ Select(B, cols c)
    Select(A op B) -> Select(A) op Select(B)        (index part only)
    Value predicates do NOT commute with matmul/add and stay in place.
    """
    def rule(n: Plan) -> Plan:
        if not isinstance(n, Select):
            return n
        pr = n.pred
        c = n.children[0]
        if pr.has_value():
            return n  # value predicate pinned here
        if isinstance(c, MatMul):
            l, r = c.children
            if pr.row_lo is not None or pr.row_hi is not None:
                l = Select(l, SelectPredicate(row_lo=pr.row_lo,
                                              row_hi=pr.row_hi))
                l = rule(l)
            if pr.col_lo is not None or pr.col_hi is not None:
                r = Select(r, SelectPredicate(col_lo=pr.col_lo,
                                              col_hi=pr.col_hi))
                r = rule(r)
            return MatMul(l, r)
        if isinstance(c, Ewise) and c.op in ("add", "sub", "mul", "min", "max"):
            l = rule(Select(c.children[0], pr))
            r = rule(Select(c.children[1], pr))
            return Ewise(l, r, c.op)
        if isinstance(c, ScalarOp) and not (c.op == "add" and c.s != 0.0):
            return ScalarOp(rule(Select(c.children[0], pr)), c.op, c.s)
        if isinstance(c, Select):
            # merge two index selections (intersection of intervals)
            q = c.pred
            if not q.has_value():
                def mn(a, b):
                    return b if a is None else (a if b is None else min(a, b))

                def mx(a, b):
                    return b if a is None else (a if b is None else max(a, b))
                merged = SelectPredicate(
                    row_lo=mx(pr.row_lo, q.row_lo),
                    row_hi=mn(pr.row_hi, q.row_hi),
                    col_lo=mx(pr.col_lo, q.col_lo),
                    col_hi=mn(pr.col_hi, q.col_hi))
                return rule(Select(c.children[0], merged))
        return n
    return _rewrite_bottom_up(p, rule)


# --------------------------------------------------------------------------
# pass: aggregate push-down through matmul
# --------------------------------------------------------------------------

def push_agg(p: Plan) -> Plan:
    """Rewrite linear aggregations of matrix products so the product is
    never materialized - MatRel's joint LA/relational optimization
    (SURVEY.md §2.1 aggregations optimized with the algebra; the classic
    SystemML/MatRel rewrite family). O(m*k*n) -> O(m*k + k*n):

      sum(A @ B)      = colsums(A) @ rowsums(B)      (1x1)
      avg(A @ B)      = sum(A @ B) / (m*n)
      trace(A @ B)    = sum(A * B^T)                 (elementwise)
      rowsums(A @ B)  = A @ rowsums(B)               (matvec)
      colsums(A @ B)  = colsums(A) @ B
      row/col avgs    = the sums scaled by 1/k-extent

    plus the linear-aggregate distributions that skip materializing the
    operand: agg(X^T) (axis swap / invariance), sum(s*X) = s*sum(X),
    sum(X+s) = sum(X) + s*extent, sum/avg/trace(A +- B) distribute.

    max/min/count/nnz/norm_fro do NOT commute with @ and stay in place.
    """
    def rule(n: Plan) -> Plan:
        if not isinstance(n, Agg):
            return n
        c = n.children[0]
        # ---- through transpose: full reductions are transpose-invariant;
        # axis reductions swap axis (rowsums(X^T) = colsums(X)^T)
        if isinstance(c, Transpose):
            x = c.children[0]
            if n.axis is None:
                return rule(Agg(x, n.kind, None))
            if n.kind in ("sum", "avg", "min", "max", "count", "nnz"):
                inner = rule(Agg(x, n.kind, 1 - n.axis))
                return Transpose(inner)
        # ---- through scalar ops: sum(s*X) = s*sum(X);
        # sum(X + s) = sum(X) + s*extent (the shifted matrix is never
        # materialized); avg analogous
        if isinstance(c, ScalarOp) and n.kind in ("sum", "avg", "trace"):
            x = c.children[0]
            if c.op == "mul":
                return ScalarOp(rule(Agg(x, n.kind, n.axis)), "mul", c.s)
            if c.op == "add" and n.kind in ("sum", "avg"):
                if n.kind == "avg":
                    return ScalarOp(rule(Agg(x, "avg", n.axis)), "add", c.s)
                extent = (x.nrows * x.ncols if n.axis is None
                          else (x.ncols if n.axis == 1 else x.nrows))
                return ScalarOp(rule(Agg(x, "sum", n.axis)), "add",
                                c.s * extent)
        # ---- through elementwise add/sub: linear aggregates distribute
        if isinstance(c, Ewise) and c.op in ("add", "sub") \
                and n.kind in ("sum", "avg", "trace"):
            l = rule(Agg(c.children[0], n.kind, n.axis))
            r = rule(Agg(c.children[1], n.kind, n.axis))
            return Ewise(l, r, c.op)
        if not isinstance(c, MatMul):
            return n
        mm = c
        a, b = mm.children
        if n.kind == "sum" and n.axis is None:
            return Agg(MatMul(Agg(a, "sum", 0), Agg(b, "sum", 1)),
                       "sum", None)
        if n.kind == "avg" and n.axis is None:
            inner = rule(Agg(mm, "sum", None))
            return ScalarOp(inner, "mul", 1.0 / (mm.nrows * mm.ncols))
        if n.kind == "trace":
            # A (m x k) @ B (k x m): tr = sum over A .* B^T
            return Agg(Ewise(a, Transpose(b), "mul"), "sum", None)
        if n.kind == "sum" and n.axis == 1:
            return MatMul(a, Agg(b, "sum", 1))
        if n.kind == "sum" and n.axis == 0:
            return MatMul(Agg(a, "sum", 0), b)
        if n.kind == "avg" and n.axis == 1:
            return ScalarOp(MatMul(a, Agg(b, "sum", 1)), "mul",
                            1.0 / mm.ncols)
        if n.kind == "avg" and n.axis == 0:
            return ScalarOp(MatMul(Agg(a, "sum", 0), b), "mul",
                            1.0 / mm.nrows)
        return n
    return _rewrite_bottom_up(p, rule)


# --------------------------------------------------------------------------
# pass: matrix-chain reordering (sparsity-aware DP)
# --------------------------------------------------------------------------

def _mm_cost(l: Plan, r: Plan) -> float:
    """Cost of one multiply (p x q) @ (q x r): flops scaled by operand
    densities (sparsity-aware, SURVEY.md §2.1 'Optimizer: matrix-chain
    ordering' - nnz-estimating cost, not just dims)."""
    p_, q_, r_ = l.nrows, l.ncols, r.ncols
    return 2.0 * p_ * q_ * r_ * max(l.density, 1e-6) * max(r.density, 1e-6)


def _collect_chain(n: Plan) -> List[Plan]:
    if isinstance(n, MatMul):
        return _collect_chain(n.children[0]) + _collect_chain(n.children[1])
    return [n]


def reorder_chains(p: Plan) -> Plan:
    def rule(n: Plan) -> Plan:
        if not isinstance(n, MatMul):
            return n
        terms = _collect_chain(n)
        if len(terms) <= 2:
            return n
        # classic chain DP over the (already-optimized) terms
        m = len(terms)
        cost: Dict[Tuple[int, int], float] = {}
        split: Dict[Tuple[int, int], int] = {}
        node: Dict[Tuple[int, int], Plan] = {}
        for i in range(m):
            cost[(i, i)] = 0.0
            node[(i, i)] = terms[i]
        for span in range(2, m + 1):
            for i in range(0, m - span + 1):
                j = i + span - 1
                best, bestk = None, None
                for k in range(i, j):
                    c = (cost[(i, k)] + cost[(k + 1, j)]
                         + _mm_cost(node[(i, k)], node[(k + 1, j)]))
                    if best is None or c < best:
                        best, bestk = c, k
                cost[(i, j)] = best
                split[(i, j)] = bestk
                node[(i, j)] = MatMul(node[(i, bestk)],
                                      node[(bestk + 1, j)])
        return node[(0, m - 1)]
    return _rewrite_bottom_up(p, rule)


# --------------------------------------------------------------------------
# pass: common-subexpression elimination (hash-consing)
# --------------------------------------------------------------------------

def cse(p: Plan) -> Plan:
    """Replace structurally identical subtrees by a single shared node.
    The executor caches results per node object, so sharing a node IS
    result reuse (reference: CSE detection & reuse)."""
    seen: Dict[tuple, Plan] = {}

    def walk(n: Plan) -> Plan:
        ch = [walk(c) for c in n.children]
        if any(a is not b for a, b in zip(ch, n.children)):
            n = n.with_children(ch)
        k = n.key()
        if k in seen:
            return seen[k]
        seen[k] = n
        return n
    return walk(p)


# --------------------------------------------------------------------------
# batch driver
# --------------------------------------------------------------------------

def optimize(p: Plan) -> Plan:
    cfg = get_config()
    p = lower_matpower(p)
    if cfg.enable_transpose_pushdown:
        p = push_transpose(p)
    if cfg.enable_scalar_folding:
        p = fold_scalars(p)
    p = push_select(p)
    if cfg.enable_agg_pushdown:
        p = push_agg(p)
    if cfg.enable_chain_reorder:
        p = reorder_chains(p)
    if cfg.enable_scalar_folding:
        p = fold_scalars(p)  # chain reorder may surface new folds
    if cfg.enable_cse:
        p = cse(p)
    return p
