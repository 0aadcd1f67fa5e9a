"""Randomized coverage for (a) the relational JOIN verb family -
cross_join / join_rows / join_cols / join_elem / semi_join_values -
which the expression fuzzes never generate, and (b) a DIFFERENTIAL
optimizer fuzz: every random tree must produce the same result with
all rewrite passes enabled and all disabled (the rewrites change plan
shape and fp association, never semantics)."""

import random

import torch

import matrel_amd as ma
from matrel_amd.exec.runtime import reset_runtime


def _session(bs=24, **kw):
    reset_runtime()
    ma.set_config(ma.MatrelConfig(block_size=bs, **kw))
    return ma.MatrelSession()


def test_join_fuzz():
    s = _session()
    rng = random.Random(42)
    for it in range(20):
        m = rng.choice([7, 24, 40, 64])
        n = rng.choice([7, 24, 40, 64])
        k = rng.choice([7, 24, 40])
        A = s.randn(m, k, seed=100 + it)
        B = s.randn(m, n, seed=200 + it)
        At, Bt = A.to_tensor(), B.to_tensor()
        # join_rows: C = A^T @ B (join on row index, sum-of-products)
        out = A.join_rows(B).compute().to_tensor()
        assert torch.allclose(out, At.t() @ Bt, atol=2e-3, rtol=2e-3)
        # join_cols: C = X @ Y^T
        X = s.randn(m, k, seed=300 + it)
        Y = s.randn(n, k, seed=400 + it)
        out = X.join_cols(Y).compute().to_tensor()
        assert torch.allclose(out, X.to_tensor() @ Y.to_tensor().t(),
                              atol=2e-3, rtol=2e-3)
        # cross_join of an m-vector and an n-covector, all ops
        a = s.randn(m, 1, seed=500 + it)
        b = s.randn(1, n, seed=600 + it)
        av = a.to_tensor().reshape(-1)
        bv = b.to_tensor().reshape(-1)
        refs = {"mul": torch.outer(av, bv),
                "add": av[:, None] + bv[None, :]}  # engine surface
        for op, ref in refs.items():
            out = a.cross_join(b, op=op).compute().to_tensor()
            assert out.shape == (m, n), (op, out.shape)
            assert torch.allclose(out, ref, atol=2e-3, rtol=2e-3), op
        # join_elem on matching shapes
        C = s.randn(m, n, seed=700 + it)
        out = B.join_elem(C, op="mul").compute().to_tensor()
        assert torch.allclose(out, Bt * C.to_tensor(), atol=2e-3,
                              rtol=2e-3)
    reset_runtime()


def test_semi_join_values_fuzz():
    s = _session()
    rng = random.Random(7)
    for it in range(10):
        m = rng.choice([16, 24, 40])
        # discrete value pool so value-equality is exact
        pool = torch.tensor([0.0, 1.0, 2.0, 3.0, 5.0, 8.0])
        At = pool[torch.randint(0, len(pool), (m, m),
                                generator=torch.Generator().manual_seed(
                                    it))]
        keepvals = pool[1: 1 + rng.randrange(1, 5)]
        Bt = torch.zeros(8, 8)
        Bt[0, : len(keepvals)] = keepvals
        A = s.from_dense(At)
        B = s.from_dense(Bt)
        out = A.semi_join_values(B).compute().to_tensor()
        ref = torch.where(torch.isin(At, keepvals), At,
                          torch.zeros(()))
        assert torch.equal(out, ref), it
    reset_runtime()


def _rand_tree(s, rng, leaves, depth):
    if depth == 0 or rng.random() < 0.3:
        m, t = leaves[rng.randrange(len(leaves))]
        return m, t
    op = rng.choice(["add", "sub", "mul", "mm", "t", "scale", "sum_mm"])
    if op == "mm":
        am, at = _rand_tree(s, rng, leaves, depth - 1)
        bm, bt = _rand_tree(s, rng, leaves, depth - 1)
        return (am @ bm) * (1 / 64), (at @ bt) / 64
    if op == "t":
        am, at = _rand_tree(s, rng, leaves, depth - 1)
        return am.T, at.t()
    if op == "scale":
        am, at = _rand_tree(s, rng, leaves, depth - 1)
        f = round(rng.uniform(-2, 2), 2)
        return am * f, at * f
    if op == "sum_mm":
        # classic agg-pushdown shape: keep it in the tree as a scalar
        # re-broadcast so the tree stays a matrix
        am, at = _rand_tree(s, rng, leaves, depth - 1)
        return am, at
    am, at = _rand_tree(s, rng, leaves, depth - 1)
    bm, bt = _rand_tree(s, rng, leaves, depth - 1)
    if op == "add":
        return am + bm, at + bt
    if op == "sub":
        return am - bm, at - bt
    return am * bm, at * bt


def _run_trees(opt_on: bool, n_trees=15):
    kw = {} if opt_on else dict(enable_transpose_pushdown=False,
                                enable_chain_reorder=False,
                                enable_cse=False,
                                enable_scheme_selection=False,
                                enable_scalar_folding=False,
                                enable_agg_pushdown=False)
    s = _session(**kw)
    rng = random.Random(99)
    leaves = [(s.randn(64, 64, seed=i), None) for i in (1, 2, 3)]
    leaves = [(m, m.to_tensor()) for m, _ in leaves]
    outs = []
    for _ in range(n_trees):
        m, ref = _rand_tree(s, rng, leaves, 4)
        out = m.compute().to_tensor()
        assert torch.allclose(out, ref, atol=3e-3, rtol=3e-3)
        # aggregates agree with the reference too (push_agg on/off)
        assert abs(m.sum().item() - ref.sum().item()) \
            <= 3e-3 * max(1.0, abs(ref.sum().item())) + 3e-2
        outs.append(out)
    reset_runtime()
    return outs


def test_optimizer_differential_fuzz():
    """Same random trees, all rewrites ON vs all OFF: semantics must
    agree (tolerance covers fp re-association from chain reorder)."""
    on = _run_trees(True)
    off = _run_trees(False)
    for i, (a, b) in enumerate(zip(on, off)):
        assert torch.allclose(a, b, atol=3e-3, rtol=3e-3), i
