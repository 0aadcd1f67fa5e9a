"""Relational layer tests: joins on indices/values, min/max combiners.
Reference: MatRel's relational operations over matrix data
(cross-product join, joins on indices/values)."""

import torch


def dense(m):
    return m.to_tensor()


def test_join_elem_combiners(session):
    A = session.randn(80, 60, seed=1)
    B = session.randn(80, 60, seed=2)
    a, b = dense(A), dense(B)
    assert torch.allclose(dense(A.join_elem(B, "mul").compute()), a * b,
                          atol=1e-5)
    assert torch.allclose(dense(A.elem_min(B).compute()),
                          torch.minimum(a, b), atol=1e-6)
    assert torch.allclose(dense(A.elem_max(B).compute()),
                          torch.maximum(a, b), atol=1e-6)


def test_join_rows_is_gram(session):
    """Row-index join with multiply-sum combiner == A^T B."""
    A = session.randn(100, 40, seed=3)
    B = session.randn(100, 50, seed=4)
    out = A.join_rows(B).compute()
    ref = dense(A).t() @ dense(B)
    assert torch.allclose(dense(out), ref, atol=1e-3)


def test_join_cols(session):
    A = session.randn(60, 100, seed=5)
    B = session.randn(70, 100, seed=6)
    out = A.join_cols(B).compute()
    ref = dense(A) @ dense(B).t()
    assert torch.allclose(dense(out), ref, atol=1e-3)


def test_semi_join_values(session):
    # A has integer-ish values so exact matches exist
    t = torch.randint(0, 5, (60, 60)).float()
    A = session.from_dense(t)
    bvals = torch.tensor([[1.0, 3.0]])
    B = session.from_dense(bvals)
    out = A.semi_join_values(B).compute()
    ref = torch.where((t == 1) | (t == 3), t, torch.zeros(()))
    assert torch.allclose(dense(out), ref, atol=0)


def test_semi_join_block_sparse(session):
    A = session.block_sparse_random(150, 150, 0.4, seed=7)
    vals = dense(A)
    B = session.from_dense(vals[:5, :5].contiguous())
    out = A.semi_join_values(B).compute()
    needles = torch.unique(vals[:5, :5][vals[:5, :5] != 0])
    ref = torch.where(torch.isin(vals, needles), vals, torch.zeros(()))
    assert torch.allclose(dense(out), ref, atol=0)


def test_cross_join_then_aggregate(session):
    """Cross-product join feeding an aggregation (reference workload
    shape: join + aggregate optimized jointly)."""
    u = session.randn(50, 1, seed=8)
    v = session.randn(1, 40, seed=9)
    total = u.cross_join(v, "mul").sum().item()
    ref = (dense(u).reshape(-1).sum() * dense(v).reshape(-1).sum()).item()
    assert abs(total - ref) < 1e-2
