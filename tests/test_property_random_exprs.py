"""Property-based testing: random expression trees evaluated by the full
engine (optimizer + physical planner + executor) must match a plain torch
evaluation of the same tree.

This is the strongest correctness net for optimizer rewrites: every
rewrite (transpose pushdown, chain DP, CSE, select pushdown, scheme
assignment) must be semantics-preserving on ARBITRARY trees, not just the
hand-written cases.
"""

import pytest
import torch
import os

from hypothesis import given, settings, strategies as st

import matrel_amd as ma
from matrel_amd.exec.runtime import reset_runtime

N = 48  # square operand side; block_size 20 -> 3x3 ragged grid


@pytest.fixture(scope="module", autouse=True)
def _session():
    reset_runtime()
    ma.set_config(ma.MatrelConfig(block_size=20))
    s = ma.MatrelSession()
    yield s
    reset_runtime()


def _leaves(s):
    torch.manual_seed(0)
    mats = []
    for seed in (1, 2, 3):
        m = s.randn(N, N, seed=seed)
        mats.append((m, m.to_tensor()))
    sp = s.block_sparse_random(N, N, 0.5, seed=4)
    mats.append((sp, sp.to_tensor()))
    return mats


# expression tree encoded as a nested spec the strategy builds
op_leaf = st.integers(min_value=0, max_value=3).map(lambda i: ("leaf", i))


def _expr(depth):
    if depth == 0:
        return op_leaf
    sub = _expr(depth - 1)
    return st.one_of(
        op_leaf,
        st.tuples(st.just("add"), sub, sub),
        st.tuples(st.just("sub"), sub, sub),
        st.tuples(st.just("mul"), sub, sub),
        st.tuples(st.just("min"), sub, sub),
        st.tuples(st.just("max"), sub, sub),
        st.tuples(st.just("mm"), sub, sub),
        st.tuples(st.just("t"), sub),
        st.tuples(st.just("scale"), sub,
                  st.floats(-2.0, 2.0, allow_nan=False)),
        st.tuples(st.just("shift"), sub,
                  st.floats(-2.0, 2.0, allow_nan=False)),
        st.tuples(st.just("sel"), sub, st.integers(0, N // 2),
                  st.integers(N // 2, N)),
    )


def _build(spec, leaves):
    kind = spec[0]
    if kind == "leaf":
        m, t = leaves[spec[1]]
        return m, t.clone()
    if kind in ("add", "sub", "mul", "min", "max", "mm"):
        am, at = _build(spec[1], leaves)
        bm, bt = _build(spec[2], leaves)
        if kind == "add":
            return am + bm, at + bt
        if kind == "sub":
            return am - bm, at - bt
        if kind == "mul":
            return am * bm, at * bt
        if kind == "min":
            return am.elem_min(bm), torch.minimum(at, bt)
        if kind == "max":
            return am.elem_max(bm), torch.maximum(at, bt)
        # mm: rescale to keep magnitudes bounded over deep trees
        return (am @ bm) * (1.0 / N), (at @ bt) / N
    if kind == "t":
        am, at = _build(spec[1], leaves)
        return am.T, at.t()
    if kind == "scale":
        am, at = _build(spec[1], leaves)
        return am * spec[2], at * spec[2]
    if kind == "shift":
        am, at = _build(spec[1], leaves)
        return am + spec[2], at + spec[2]
    if kind == "sel":
        am, at = _build(spec[1], leaves)
        lo, hi = spec[2], spec[3]
        ref = torch.zeros_like(at)
        ref[lo:hi] = at[lo:hi]
        return am.select_rows(lo, hi), ref
    raise ValueError(kind)


@settings(max_examples=int(os.environ.get("MATREL_HYP_EXAMPLES", "150")),
          deadline=None, derandomize=True)
@given(spec=_expr(4))
def test_random_expression_matches_torch(spec, _session):
    leaves = _leaves(_session)
    m, ref = _build(spec, leaves)
    out = m.compute().to_tensor()
    assert torch.allclose(out, ref, atol=1e-3, rtol=1e-3), \
        f"mismatch for {spec}: max err " \
        f"{(out - ref).abs().max().item():.3e}"


@settings(max_examples=60, deadline=None, derandomize=True)
@given(spec=_expr(3),
       agg=st.sampled_from(["sum", "max", "min", "row_sums", "col_sums"]))
def test_random_expression_aggregates(spec, agg, _session):
    leaves = _leaves(_session)
    m, ref = _build(spec, leaves)
    if agg == "sum":
        assert abs(m.sum().item() - ref.sum().item()) < 1e-2 * N
    elif agg == "max":
        assert abs(m.max().item() - ref.max().item()) < 1e-4
    elif agg == "min":
        assert abs(m.min().item() - ref.min().item()) < 1e-4
    elif agg == "row_sums":
        assert torch.allclose(m.row_sums().to_tensor().reshape(-1),
                              ref.sum(dim=1), atol=1e-2)
    else:
        assert torch.allclose(m.col_sums().to_tensor().reshape(-1),
                              ref.sum(dim=0), atol=1e-2)
