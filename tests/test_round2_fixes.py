"""Round-2 regression tests for single-process correctness fixes:

* Cholesky fast path in Solve/Inverse is symmetry-gated (ADVICE.md r1
  high: a non-symmetric matrix with an SPD lower triangle used to get
  silently wrong results).
* Solve/Inverse refuse to gather operands above MATREL_SOLVE_MAX_SIDE
  (VERDICT r1 weak #6).
* Direct `m.blocks[...] = t` writes invalidate cached presence/nnz maps
  (stale nnz map breaks the SUMMA CSR-wire sizing).
"""

import pytest
import torch

import matrel_amd as ma


@pytest.fixture(autouse=True)
def _small_blocks():
    ma.set_config(ma.MatrelConfig(block_size=32))
    yield


def _nonsym_spd_lower(n: int = 48) -> torch.Tensor:
    """Non-symmetric matrix whose lower triangle would pass Cholesky."""
    g = torch.Generator().manual_seed(7)
    r = torch.randn(n, n, generator=g)
    spd = r @ r.t() + n * torch.eye(n)  # SPD
    a = spd.clone()
    # replace the strict upper triangle -> non-symmetric, still invertible
    a += torch.triu(torch.randn(n, n, generator=g), diagonal=1)
    assert not torch.equal(a, a.t())
    return a


def test_inverse_nonsymmetric_spd_lower():
    s = ma.MatrelSession()
    a = _nonsym_spd_lower()
    A = s.from_dense(a)
    inv = A.inverse().compute().to_tensor()
    ref = torch.linalg.inv(a)
    assert torch.allclose(inv, ref, atol=1e-3), \
        (inv - ref).abs().max().item()


def test_solve_nonsymmetric_spd_lower():
    s = ma.MatrelSession()
    a = _nonsym_spd_lower()
    b = torch.randn(48, 3, generator=torch.Generator().manual_seed(9))
    x = s.from_dense(a).solve(s.from_dense(b)).compute().to_tensor()
    ref = torch.linalg.solve(a, b)
    assert torch.allclose(x, ref, atol=1e-3), \
        (x - ref).abs().max().item()


def test_inverse_spd_still_works():
    s = ma.MatrelSession()
    g = torch.Generator().manual_seed(3)
    r = torch.randn(48, 48, generator=g)
    spd = r @ r.t() + 48 * torch.eye(48)
    inv = s.from_dense(spd).inverse().compute().to_tensor()
    assert torch.allclose(inv, torch.linalg.inv(spd), atol=1e-3)


def test_solve_max_side_guard():
    cfg = ma.MatrelConfig(block_size=32)
    cfg.solve_max_side = 64
    s = ma.MatrelSession(config=cfg)
    A = s.randn(128, 128, seed=1)
    with pytest.raises(ValueError, match="MATREL_SOLVE_MAX_SIDE"):
        A.inverse().compute()
    with pytest.raises(ValueError, match="MATREL_SOLVE_MAX_SIDE"):
        A.solve(s.randn(128, 1, seed=2)).compute()


def test_stats_invalidated_on_direct_block_write():
    from matrel_amd.blocks.matrix import BlockMatrix
    m = BlockMatrix.randn(64, 64, seed=5)  # 2x2 blocks of 32
    nm0 = m.nnz_map().clone()
    pm0 = m.presence().clone()
    assert pm0.sum() == 4
    # direct dict write (NOT set_block): sparsify one block
    sp = torch.zeros(32, 32)
    sp[0, 0] = 1.0
    m.blocks[(0, 0)] = sp.to_sparse_csr()
    nm1 = m.nnz_map()
    assert nm1[0, 0] == 1, (nm0[0, 0].item(), nm1[0, 0].item())
    # direct delete: presence must drop
    del m.blocks[(1, 1)]
    assert m.presence().sum() == 3


def test_maxmin_single_process_all_negative():
    """Single-process sanity companion to the world-2 regression."""
    s = ma.MatrelSession()
    a = -torch.rand(64, 64) - 1.0
    A = s.from_dense(a)
    assert abs(A.max().item() - a.max().item()) < 1e-5
    assert abs(A.min().item() - a.min().item()) < 1e-5


def test_blockmatrix_dies_by_refcount():
    """BlockMatrix must be freed by REFCOUNT, not the cycle collector:
    a strong blocks->owner backref once made multi-hundred-GB result
    matrices linger until gc intervals (OOM at the near-capacity
    163840^2 bench)."""
    import gc
    import weakref
    from matrel_amd.blocks.matrix import BlockMatrix
    gc.disable()
    try:
        m = BlockMatrix.randn(64, 64, seed=9)
        m.presence()  # populate caches
        m.blocks[(0, 0)] = torch.zeros(32, 32)  # exercise the store
        r = weakref.ref(m)
        del m
        assert r() is None, "BlockMatrix survived del (reference cycle)"
    finally:
        gc.enable()


def test_no_result_accumulation_across_steps():
    """Iterative workloads (the bench, power iteration) must not retain
    discarded step outputs: every previous result's BlockMatrix dies by
    refcount before the next step completes."""
    import gc
    import weakref
    s = ma.MatrelSession()
    A = s.randn(128, 128, seed=21)
    B = s.randn(128, 128, seed=22)
    gc.disable()
    try:
        dead = []
        prev = None
        for _ in range(5):
            C = (A @ B).compute()
            if prev is not None:
                dead.append(prev)
            prev = weakref.ref(C.plan.data)
        del C
        dead.append(prev)
        alive = [r for r in dead if r() is not None]
        assert not alive, f"{len(alive)} step outputs retained"
    finally:
        gc.enable()


def test_spgemm_declines_in_deterministic_mode():
    """deterministic=True must route sparse x sparse away from the
    LDS-atomic SpGEMM kernel (lane-order-dependent sums) while staying
    correct through the densified path."""
    cfg = ma.MatrelConfig(block_size=32)
    cfg.deterministic = True
    s = ma.MatrelSession(config=cfg)
    A = s.sparse_random(96, 96, 0.05, seed=31)
    B = s.sparse_random(96, 64, 0.05, seed=32)
    C = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)
