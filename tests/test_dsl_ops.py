"""DSL + executor correctness vs dense torch references (CPU).

Mirrors the reference's unit-test strategy: per-op
correctness against a small dense reference, here torch fp32.
Block size 48 with non-multiple matrix dims exercises ragged edges and
multi-block paths.
"""

import math

import pytest
import torch


def dense(m):
    return m.to_tensor()


@pytest.fixture()
def abc(session):
    A = session.randn(100, 70, seed=1)
    B = session.randn(70, 90, seed=2)
    C = session.randn(100, 70, seed=3)
    return A, B, C


def test_matmul(session, abc):
    A, B, _ = abc
    out = (A @ B).compute()
    ref = dense(A) @ dense(B)
    assert torch.allclose(dense(out), ref, atol=1e-4)


def test_matmul_chain(session):
    A = session.randn(60, 50, seed=1)
    B = session.randn(50, 40, seed=2)
    C = session.randn(40, 30, seed=3)
    out = (A @ B @ C).compute()
    ref = dense(A) @ dense(B) @ dense(C)
    assert torch.allclose(dense(out), ref, atol=1e-3)


def test_add_sub_mul_div(session, abc):
    A, _, C = abc
    for op, f in [("+", lambda a, b: a + b), ("-", lambda a, b: a - b),
                  ("*", lambda a, b: a * b)]:
        out = f(A, C).compute()
        ref = f(dense(A), dense(C))
        assert torch.allclose(dense(out), ref, atol=1e-5), op
    # division: avoid zeros in denominator
    D = (C * C + 1.0).compute()
    out = (A / D).compute()
    ref = dense(A) / dense(D)
    assert torch.allclose(dense(out), ref, atol=1e-5)


def test_scalar_ops(session, abc):
    A, _, _ = abc
    out = (2.5 * A + 1.0 - 0.5).compute()
    ref = 2.5 * dense(A) + 0.5
    assert torch.allclose(dense(out), ref, atol=1e-5)
    out2 = (-A).compute()
    assert torch.allclose(dense(out2), -dense(A), atol=1e-6)
    out3 = (1.0 - A).compute()
    assert torch.allclose(dense(out3), 1.0 - dense(A), atol=1e-5)


def test_elem_pow(session):
    A = session.rand(50, 60, seed=4)  # positive entries
    out = (A ** 2).compute()
    assert torch.allclose(dense(out), dense(A) ** 2, atol=1e-5)


def test_transpose(session, abc):
    A, B, _ = abc
    out = A.T.compute()
    assert torch.allclose(dense(out), dense(A).t(), atol=0)
    # (AB)^T == B^T A^T (optimizer rewrites this form)
    out2 = (A @ B).T.compute()
    ref = (dense(A) @ dense(B)).t()
    assert torch.allclose(dense(out2), ref, atol=1e-4)


def test_matpow(session):
    A = session.randn(64, 64, seed=5) * 0.1
    out = A.matpow(3).compute()
    a = dense(A)
    assert torch.allclose(dense(out), a @ a @ a, atol=1e-4)


def test_aggregates_full(session, abc):
    A, _, _ = abc
    a = dense(A)
    assert math.isclose(A.sum().item(), a.sum().item(), rel_tol=1e-4)
    assert math.isclose(A.avg().item(), a.mean().item(), rel_tol=1e-4)
    assert math.isclose(A.max().item(), a.max().item(), rel_tol=1e-5)
    assert math.isclose(A.min().item(), a.min().item(), rel_tol=1e-5)
    assert math.isclose(A.norm_fro().item(),
                        torch.linalg.norm(a).item(), rel_tol=1e-4)
    assert A.count().item() == float((a != 0).sum().item())


def test_trace(session):
    A = session.randn(100, 100, seed=6)
    assert math.isclose(A.trace().item(),
                        torch.trace(dense(A)).item(), rel_tol=1e-4)


def test_row_col_aggregates(session, abc):
    A, _, _ = abc
    a = dense(A)
    assert torch.allclose(dense(A.row_sums()).reshape(-1), a.sum(dim=1),
                          atol=1e-4)
    assert torch.allclose(dense(A.col_sums()).reshape(-1), a.sum(dim=0),
                          atol=1e-4)
    assert torch.allclose(dense(A.row_maxs()).reshape(-1),
                          a.max(dim=1).values, atol=1e-6)
    assert torch.allclose(dense(A.col_mins()).reshape(-1),
                          a.min(dim=0).values, atol=1e-6)
    assert torch.allclose(dense(A.row_avgs()).reshape(-1), a.mean(dim=1),
                          atol=1e-5)


def test_select_index(session, abc):
    A, _, _ = abc
    out = A.select(rows=(10, 60), cols=(5, 50)).compute()
    ref = torch.zeros_like(dense(A))
    ref[10:60, 5:50] = dense(A)[10:60, 5:50]
    assert torch.allclose(dense(out), ref, atol=0)


def test_select_value(session, abc):
    A, _, _ = abc
    out = A.select_values(0.0, 1.0).compute()
    a = dense(A)
    ref = torch.where((a >= 0) & (a <= 1), a, torch.zeros(()))
    assert torch.allclose(dense(out), ref, atol=0)
    out2 = A.select_values(0.0, 1.0, inside=False).compute()
    ref2 = torch.where((a >= 0) & (a <= 1), torch.zeros(()), a)
    assert torch.allclose(dense(out2), ref2, atol=0)


def test_select_then_aggregate(session, abc):
    """The reference's A.select(rowIdx < n).rowSum() path."""
    A, _, _ = abc
    out = A.select_rows(0, 50).row_sums().compute()
    a = dense(A)
    ref = torch.zeros(a.shape[0])
    ref[:50] = a[:50].sum(dim=1)
    assert torch.allclose(dense(out).reshape(-1), ref, atol=1e-4)


def test_cross_join(session):
    u = session.randn(80, 1, seed=7)
    v = session.randn(1, 60, seed=8)
    out = u.cross_join(v, op="mul").compute()
    ref = torch.outer(dense(u).reshape(-1), dense(v).reshape(-1))
    assert torch.allclose(dense(out), ref, atol=1e-5)
    out2 = u.cross_join(v, op="add").compute()
    ref2 = dense(u).reshape(-1, 1) + dense(v).reshape(1, -1)
    assert torch.allclose(dense(out2), ref2, atol=1e-5)


def test_solve_and_inverse(session):
    X = session.randn(120, 30, seed=9)
    # SPD system: (X^T X) w = X^T y
    y = session.randn(120, 1, seed=10)
    XtX = (X.T @ X).compute()
    Xty = (X.T @ y).compute()
    w = XtX.solve(Xty).compute()
    ref = torch.linalg.solve(dense(XtX).double(), dense(Xty).double())
    assert torch.allclose(dense(w).double(), ref, atol=1e-3)
    inv = XtX.inverse().compute()
    refinv = torch.linalg.inv(dense(XtX).double())
    assert torch.allclose(dense(inv).double(), refinv, atol=1e-3)


def test_ols_pipeline(session):
    """BASELINE.json config 5 shape: w = (X^T X)^-1 X^T y."""
    X = session.randn(200, 25, seed=11)
    y = session.randn(200, 1, seed=12)
    w = (X.T @ X).inverse() @ (X.T @ y)
    out = w.compute()
    x, yy = dense(X).double(), dense(y).double()
    ref = torch.linalg.lstsq(x, yy).solution
    assert torch.allclose(dense(out).double(), ref, atol=1e-2)


def test_block_sparse(session):
    A = session.block_sparse_random(200, 200, 0.3, seed=13)
    B = session.randn(200, 100, seed=14)
    out = (A @ B).compute()
    ref = dense(A) @ dense(B)
    assert torch.allclose(dense(out), ref, atol=1e-4)
    # sum skips absent blocks but matches dense
    assert math.isclose(A.sum().item(), dense(A).sum().item(),
                        rel_tol=1e-3, abs_tol=1e-3)


def test_reflected_scalar_ops(session):
    A = (session.rand(64, 64, seed=9) + 0.5).compute()
    t = A.to_tensor()
    for expr, ref in [
        (2.0 - A, 2.0 - t), (2.0 + A, 2.0 + t), (2.0 * A, 2.0 * t),
        (-A, -t), (A / 2.0, t / 2.0), (2.0 / A, 2.0 / t),
        (A ** 2, t ** 2),
    ]:
        assert torch.allclose(expr.to_tensor(), ref, atol=1e-4)


def test_explain(session):
    X = session.randn(128, 16, seed=21)
    y = session.randn(128, 1, seed=22)
    txt = ((X.T @ X).inverse() @ (X.T @ y)).explain(verbose=False)
    assert "== logical plan ==" in txt
    assert "== optimized plan ==" in txt
    assert "algo=" in txt
    assert "cse-shared" in txt  # the shared X^T must be marked


def test_row_col_norms(session):
    A = session.randn(96, 64, seed=31)
    t = A.to_tensor()
    rn = A.row_norms().to_tensor().reshape(-1)
    assert torch.allclose(rn, t.norm(dim=1), atol=1e-3)
    cn = A.col_norms().to_tensor().reshape(-1)
    assert torch.allclose(cn, t.norm(dim=0), atol=1e-3)
