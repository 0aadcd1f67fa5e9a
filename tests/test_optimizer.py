"""Optimizer rewrite tests: assert plan SHAPE after rules, like the
reference's optimizer suites ("assert plan shape after rules:
transpose pushed down, chain reordered")."""

import pytest
import torch

from matrel_amd.plan import logical as L
from matrel_amd.plan.optimizer import (cse, fold_scalars, lower_matpower,
                                       optimize, push_select,
                                       push_transpose, reorder_chains)


@pytest.fixture()
def mats(session):
    A = session.randn(100, 80, seed=1)
    B = session.randn(80, 60, seed=2)
    return A, B


def count_nodes(p, cls):
    seen = set()
    n = 0

    def walk(x):
        nonlocal n
        if id(x) in seen:
            return
        seen.add(id(x))
        if isinstance(x, cls):
            n += 1
        for c in x.children:
            walk(c)
    walk(p)
    return n


def test_double_transpose_eliminated(mats):
    A, _ = mats
    p = push_transpose(A.T.T.plan)
    assert isinstance(p, L.Leaf)


def test_transpose_pushdown_matmul(mats):
    A, B = mats
    p = push_transpose((A @ B).T.plan)
    # (AB)^T -> B^T A^T : root must be MatMul, transposes at leaves
    assert isinstance(p, L.MatMul)
    assert isinstance(p.children[0], L.Transpose)
    assert isinstance(p.children[0].children[0], L.Leaf)
    assert isinstance(p.children[1], L.Transpose)


def test_transpose_pushdown_correctness(session, mats):
    A, B = mats
    ref = (A.to_tensor() @ B.to_tensor()).t()
    out = (A @ B).T.compute()
    assert torch.allclose(out.to_tensor(), ref, atol=1e-4)


def test_scalar_folding(mats):
    A, _ = mats
    expr = ((A * 2.0) * 3.0).plan
    p = fold_scalars(expr)
    assert isinstance(p, L.ScalarOp) and p.s == 6.0
    assert isinstance(p.children[0], L.Leaf)


def test_scalar_pulled_out_of_matmul(mats):
    A, B = mats
    p = fold_scalars(((A * 2.0) @ B).plan)
    assert isinstance(p, L.ScalarOp) and p.s == 2.0
    assert isinstance(p.children[0], L.MatMul)


def test_chain_reorder_small_middle(session):
    # A(100x80) @ B(80x5) @ C(5x90): right-assoc first is much cheaper
    A = session.randn(100, 80, seed=1)
    B = session.randn(80, 5, seed=2)
    C = session.randn(5, 90, seed=3)
    p = reorder_chains((A @ B @ C).plan)
    # optimal: (A@B) @ C -> cost 100*80*5 + 100*5*90 = 85k
    # vs A @ (B@C)       -> cost 80*5*90 + 100*80*90 = 756k
    assert isinstance(p, L.MatMul)
    assert isinstance(p.children[0], L.MatMul)  # (A@B) first
    res = ((A @ B) @ C).compute().to_tensor()
    res2 = (A @ (B @ C)).compute().to_tensor()
    assert torch.allclose(res, res2, atol=1e-3)


def test_chain_reorder_sparsity_aware(session):
    """A dense x (sparse @ dense): sparsity shifts the optimal split."""
    A = session.randn(100, 100, seed=1)
    S = session.block_sparse_random(100, 100, 0.05, seed=2)
    C = session.randn(100, 100, seed=3)
    p = reorder_chains((S @ A @ C).plan)
    # S is low density -> (S@A) first is the cheap split; the DP must pick
    # a left-leaning tree whose first multiply involves S.
    assert isinstance(p, L.MatMul)
    assert isinstance(p.children[0], L.MatMul)


def test_cse_shares_subtrees(mats):
    A, B = mats
    AB1 = A @ B
    AB2 = A @ B
    expr = (AB1 + AB2).plan
    p = cse(expr)
    assert p.children[0] is p.children[1]


def test_cse_execution_count(session, monkeypatch):
    """Shared subtree must execute once (reuse)."""
    from matrel_amd.exec import summa
    calls = {"n": 0}
    orig = summa.local_matmul

    def counting(a, b, out):
        calls["n"] += 1
        return orig(a, b, out)
    monkeypatch.setattr(summa, "local_matmul", counting)
    monkeypatch.setattr("matrel_amd.exec.executor.local_matmul", counting)
    A = session.randn(60, 60, seed=1)
    B = session.randn(60, 60, seed=2)
    expr = (A @ B) + (A @ B)
    expr.compute()
    assert calls["n"] == 1


def test_matpower_lowering(session):
    """A^4 lowers to 2 matmuls after CSE (square, then square the square)."""
    Sq = session.randn(64, 64, seed=5)
    lowered = cse(lower_matpower(Sq.matpow(4).plan))
    assert count_nodes(lowered, L.MatMul) == 2  # sq = A@A; sq@sq


def test_select_pushdown_through_matmul(session):
    A = session.randn(100, 80, seed=1)
    B = session.randn(80, 60, seed=2)
    expr = (A @ B).select(rows=(0, 10), cols=(0, 20)).plan
    p = push_select(expr)
    assert isinstance(p, L.MatMul)
    assert isinstance(p.children[0], L.Select)  # row select on A
    assert isinstance(p.children[1], L.Select)  # col select on B
    # correctness
    out_opt = (A @ B).select(rows=(0, 10), cols=(0, 20)).compute()
    ref = torch.zeros(100, 60)
    ab = A.to_tensor() @ B.to_tensor()
    ref[0:10, 0:20] = ab[0:10, 0:20]
    assert torch.allclose(out_opt.to_tensor(), ref, atol=1e-4)


def test_full_optimize_pipeline_correct(session):
    A = session.randn(90, 70, seed=1)
    B = session.randn(70, 50, seed=2)
    C = session.randn(50, 40, seed=3)
    expr = ((A @ B @ C).T * 2.0).T + (A @ B @ C)
    out = expr.compute().to_tensor()
    a, b, c = A.to_tensor(), B.to_tensor(), C.to_tensor()
    ref = 3.0 * (a @ b @ c)
    assert torch.allclose(out, ref, atol=1e-3)


def test_supertile_map_bijective():
    """Python replica of csrc/common.h supertile_map: must be a bijection
    id -> (tm, tn) for every guarded geometry."""
    def smap(i, ntm, ntn, SM=4, SN=8):
        per = SM * SN
        st, w = divmod(i, per)
        stm, stn = divmod(st, ntn // SN)
        return (stm * SM + w // SN, stn * SN + w % SN)

    for ntm, ntn in [(8, 8), (16, 16), (32, 32), (64, 64), (4, 8), (12, 24)]:
        seen = {smap(i, ntm, ntn) for i in range(ntm * ntn)}
        assert len(seen) == ntm * ntn
        assert all(0 <= tm < ntm and 0 <= tn < ntn for tm, tn in seen)


def test_agg_pushdown_sum_avoids_matmul(session):
    """sum(A@B) must plan as colsums(A) @ rowsums(B) - the O(n^3) product
    is never materialized (MatRel's joint LA/relational rewrite family)."""
    from matrel_amd.plan import logical as L
    from matrel_amd.plan.optimizer import optimize
    A = session.randn(96, 64, seed=1)
    B = session.randn(64, 80, seed=2)
    opt = optimize((A @ B).sum().plan)

    def mms(n, acc):
        if isinstance(n, L.MatMul):
            acc.append((n.children[0].shape, n.children[1].shape))
        for c in n.children:
            mms(c, acc)
        return acc
    shapes = mms(opt, [])
    assert all(s[0][0] == 1 or s[1][1] == 1 for s in shapes), shapes


def test_agg_pushdown_numeric(session):
    A = session.randn(96, 64, seed=3)
    B = session.randn(64, 80, seed=4)
    C = session.randn(64, 96, seed=5)
    ta, tb, tc = A.to_tensor(), B.to_tensor(), C.to_tensor()
    p = ta @ tb
    assert abs((A @ B).sum().item() - p.sum().item()) <= \
        1e-3 * abs(p.sum().item()) + 2e-2
    assert abs((A @ B).avg().item() - p.mean().item()) < 1e-3
    assert abs((A @ C).trace().item() - (ta @ tc).trace().item()) < 2e-2
    assert torch.allclose((A @ B).row_sums().to_tensor().reshape(-1),
                          p.sum(dim=1), atol=1e-2, rtol=1e-3)
    assert torch.allclose((A @ B).col_sums().to_tensor().reshape(-1),
                          p.sum(dim=0), atol=1e-2, rtol=1e-3)
    assert torch.allclose((A @ B).row_avgs().to_tensor().reshape(-1),
                          p.mean(dim=1), atol=1e-3)
    assert torch.allclose((A @ B).col_avgs().to_tensor().reshape(-1),
                          p.mean(dim=0), atol=1e-3)
    # non-linear aggregates must NOT be rewritten (max of product != any
    # aggregate of the factors)
    assert abs((A @ B).max().item() - p.max().item()) < 1e-2


def test_agg_pushdown_ablation(session):
    import matrel_amd as ma
    A = session.randn(64, 64, seed=6)
    B = session.randn(64, 64, seed=7)
    ref = (A @ B).sum().item()
    cfg = ma.get_config()
    cfg.enable_agg_pushdown = False
    try:
        off = (A @ B).sum().item()
    finally:
        cfg.enable_agg_pushdown = True
    assert abs(ref - off) <= 1e-3 * abs(off) + 2e-2


def test_agg_linear_distributions(session):
    """agg(X^T) axis swap, sum(s*X), sum(X+s), sum/trace over A+-B - the
    operand is never materialized; numerics must match torch."""
    A = session.randn(96, 64, seed=11)
    B = session.randn(96, 64, seed=12)
    ta, tb = A.to_tensor(), B.to_tensor()
    assert abs(A.T.sum().item() - ta.sum().item()) < 2e-2
    assert torch.allclose(A.T.row_sums().to_tensor().reshape(-1),
                          ta.sum(dim=0), atol=1e-2)
    assert torch.allclose(A.T.col_maxs().to_tensor().reshape(-1),
                          ta.max(dim=1).values, atol=1e-4)
    assert abs((3.0 * A).sum().item() - 3 * ta.sum().item()) < 5e-2
    assert abs((A + 2.0).sum().item() - (ta + 2).sum().item()) < 5e-1
    assert torch.allclose((A + 2.0).row_sums().to_tensor().reshape(-1),
                          (ta + 2).sum(dim=1), atol=1e-1)
    assert abs((A + 2.0).avg().item() - (ta + 2).mean().item()) < 1e-3
    assert abs((A + B).sum().item() - (ta + tb).sum().item()) < 5e-2
    assert torch.allclose((A - B).col_sums().to_tensor().reshape(-1),
                          (ta - tb).sum(dim=0), atol=2e-2)
    C = session.randn(64, 64, seed=13)
    D = session.randn(64, 64, seed=14)
    assert abs((C - D).trace().item()
               - (C.to_tensor() - D.to_tensor()).trace().item()) < 2e-2
