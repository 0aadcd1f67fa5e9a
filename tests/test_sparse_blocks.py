"""Within-block sparse (CSR) storage: auto-sparsification, SpMM path,
ops on sparse blocks (reference: per-block CSC + auto dense/sparse by nnz,
SURVEY.md §2.1)."""

import math

import pytest
import torch

from matrel_amd.ops import blockops


def dense(m):
    return m.to_tensor()


def test_sparse_random_storage(session):
    A = session.sparse_random(200, 150, 0.03, seed=1)
    bm = A.blocks
    assert all(blockops.is_sparse(t) for _, t in bm.local_items())
    d = dense(A)
    frac = (d != 0).float().mean().item()
    assert 0.01 < frac < 0.06


def test_from_dense_auto_sparsify(session):
    t = torch.zeros(100, 100)
    t[0, 0] = 1.0
    t[50, 70] = 2.0
    A = session.from_dense(t)
    bm = A.blocks
    kinds = [blockops.is_sparse(b) for _, b in bm.local_items()]
    assert any(kinds), "near-empty blocks should be CSR"
    assert torch.allclose(dense(A), t, atol=0)


def test_spmm_matches_dense(session):
    A = session.sparse_random(150, 120, 0.05, seed=2)
    B = session.randn(120, 90, seed=3)
    out = (A @ B).compute()
    ref = dense(A) @ dense(B)
    assert torch.allclose(dense(out), ref, atol=1e-3)


def test_sparse_aggregates(session):
    A = session.sparse_random(150, 150, 0.04, seed=4)
    d = dense(A)
    assert math.isclose(A.sum().item(), d.sum().item(),
                        rel_tol=1e-3, abs_tol=1e-3)
    assert A.count().item() == float((d != 0).sum().item())
    assert math.isclose(A.norm_fro().item(),
                        torch.linalg.norm(d).item(), rel_tol=1e-4)


def test_sparse_elementwise_and_transpose(session):
    A = session.sparse_random(100, 80, 0.05, seed=5)
    B = session.randn(100, 80, seed=6)
    out = (A + B).compute()
    assert torch.allclose(dense(out), dense(A) + dense(B), atol=1e-5)
    out2 = A.T.compute()
    assert torch.allclose(dense(out2), dense(A).t(), atol=0)


def test_sparse_save_load(session, tmp_path):
    A = session.sparse_random(120, 120, 0.04, seed=7)
    p = str(tmp_path / "sp")
    A.save(p)
    B = session.load(p)
    assert torch.allclose(dense(A), dense(B), atol=0)


@pytest.mark.gpu
def test_spmm_gpu_kernel():
    from matrel_amd.ops.backend import require_ext
    ext = require_ext()
    g = torch.Generator().manual_seed(8)
    a = torch.randn(300, 256, generator=g)
    a = a * (torch.rand(300, 256, generator=g) < 0.05)
    a_csr = a.to(torch.bfloat16).cuda().to_sparse_csr()
    b = torch.randn(256, 200, generator=g).to(torch.bfloat16).cuda()
    c = torch.zeros(300, 200, dtype=torch.float32, device="cuda")
    blockops.spmm_acc(c, a_csr, b)
    ref = a.to(torch.bfloat16).float() @ b.float().cpu()
    assert torch.allclose(c.cpu(), ref, atol=0.1)


def test_spgemm_cpu_fallback(session):
    """sparse x sparse block product through local_matmul (CPU path)."""
    s = session
    A = s.sparse_random(150, 120, 0.05, seed=11)
    B = s.sparse_random(120, 100, 0.05, seed=12)
    out = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(dense(out), ref, atol=1e-3)


@pytest.mark.gpu
def test_spgemm_gpu_kernel():
    """CSR x CSR HIP kernel (ds_add_f32 LDS accumulator) vs torch fp32."""
    from matrel_amd.ops.backend import require_ext
    require_ext()
    g = torch.Generator().manual_seed(13)
    for m, k, n, d in ((300, 256, 200, 0.05), (1024, 1024, 1024, 0.01),
                       (512, 384, 16384, 0.03)):
        a = torch.randn(m, k, generator=g) * (torch.rand(m, k,
                                                         generator=g) < d)
        b = torch.randn(k, n, generator=g) * (torch.rand(k, n,
                                                         generator=g) < d)
        a_csr = a.to(torch.bfloat16).cuda().to_sparse_csr()
        b_csr = b.to(torch.bfloat16).cuda().to_sparse_csr()
        c = torch.zeros(m, n, dtype=torch.float32, device="cuda")
        assert blockops.spgemm_acc(c, a_csr, b_csr)
        ref = (a.to(torch.bfloat16).float() @ b.to(torch.bfloat16).float())
        err = (c.cpu() - ref).abs().max().item()
        assert err < 0.1, (m, k, n, d, err)
    # accumulate semantics: second call adds
    assert blockops.spgemm_acc(c, a_csr, b_csr)
    err2 = (c.cpu() - 2 * ref).abs().max().item()
    assert err2 < 0.2, err2


@pytest.mark.gpu
def test_sparse_end_to_end_gpu(gpu_session):
    s = gpu_session
    A = s.sparse_random(2048, 2048, 0.02, seed=9)
    B = s.randn(2048, 1024, seed=10)
    out = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert (out.to_tensor() - ref).abs().max().item() < 0.5


def test_both_sparse_ragged_blocks():
    """Regression (cpu_fuzz_sweep): both-operands-sparse matmul with a
    RAGGED trailing block (n % bs != 0) - the spgemm fallback must add
    the narrower product into the padded accumulator slice."""
    import matrel_amd as ma
    from matrel_amd.exec.runtime import reset_runtime
    reset_runtime()
    ma.set_config(ma.MatrelConfig(block_size=24))
    s = ma.MatrelSession()
    try:
        a = s.sparse_random(64, 64, 0.05, seed=11)
        b = s.sparse_random(64, 64, 0.05, seed=12)
        out = (a @ b).compute().to_tensor()
        ref = a.to_tensor() @ b.to_tensor()
        tol = 2e-2 if out.is_cuda else 2e-3  # GPU sessions hold bf16
        assert torch.allclose(out.float(), ref.float(), atol=tol,
                              rtol=tol)
    finally:
        reset_runtime()
