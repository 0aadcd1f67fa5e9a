"""World-size-4 gloo tests: the 2x2 grid exercises BOTH non-trivial grid
dimensions (row AND column sub-groups have >1 member), matching the shape
class of the driver's 4- and 8-GPU SCALE runs (2x2 / 4x2 grids). The
world-2 suite (test_distributed.py) only covers a 2x1 grid.
"""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

WORLD = 4


def _worker(rank: int, world: int, port: int, fn_name: str, tmpdir: str):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["HSA_ENABLE_IPC_MODE_LEGACY"] = "0"
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import matrel_amd as ma
    ma.set_config(ma.MatrelConfig(block_size=32))
    globals()[fn_name](ma.MatrelSession(), tmpdir)
    dist.destroy_process_group()


def _run(fn_name: str, tmp_path) -> None:
    port = 29965 + abs(hash(fn_name)) % 30
    mp.spawn(_worker, args=(WORLD, port, fn_name, str(tmp_path)),
             nprocs=WORLD, join=True)


def _body_summa_2x2(s, tmpdir):
    # ragged edges + a k dimension that is not a multiple of the grid
    A = s.randn(160, 112, seed=11)
    B = s.randn(112, 96, seed=12)
    C = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)


def _body_chain_2x2(s, tmpdir):
    # chained multiply + transpose through the optimizer on the 2x2 grid
    A = s.randn(96, 64, seed=13)
    B = s.randn(96, 64, seed=14)
    C = (A.T @ B).compute()
    ref = A.to_tensor().t() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)


def _body_sparse_2x2(s, tmpdir):
    A = s.block_sparse_random(128, 128, 0.3, seed=15)
    B = s.randn(128, 64, seed=16)
    C = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)
    # element-sparse (CSR) operand + transpose through the full path
    E = s.sparse_random(128, 96, 0.03, seed=17)
    D = (E.T @ A).compute()
    refd = E.to_tensor().t() @ A.to_tensor()
    assert torch.allclose(D.to_tensor(), refd, atol=1e-3)


def _body_agg_2x2(s, tmpdir):
    A = s.randn(96, 96, seed=17)
    t = A.to_tensor()
    assert abs(A.sum().item() - t.sum().item()) < 1e-1
    rs = A.row_sums().to_tensor().reshape(-1)
    assert torch.allclose(rs, t.sum(dim=1), atol=1e-2)


def _body_both_sparse_2x2(s, tmpdir):
    """BOTH operands element-sparse on the 2x2 grid: unlike world 2
    (2x1 grid, where the A-panel row group has a single member), BOTH
    CSR wire directions here ride real multi-rank broadcasts - the
    variable-size idx/val buffer sizing must agree group-uniformly
    from the nnz maps on every rank."""
    from matrel_amd.blocks.matrix import BlockMatrix
    from matrel_amd.blocks.schemes import GRID
    from matrel_amd.exec import metrics
    from matrel_amd.exec.summa import summa_matmul
    metrics.reset()
    for seed in (61, 62):
        A = BlockMatrix.sparse_random(160, 128, 0.03, seed=seed)
        B = BlockMatrix.sparse_random(128, 96, 0.03, seed=seed + 10)
        out = BlockMatrix(160, 96, A.block_size, GRID, A.dtype, A.device)
        summa_matmul(A, B, out)
        ref = A.to_dense() @ B.to_dense()
        assert torch.allclose(out.to_dense(), ref, atol=1e-3), seed
    rep = metrics.report()
    assert "summa_a_bcast_csr" in rep["collectives"]
    assert "summa_b_bcast_csr" in rep["collectives"]
    # mixed sparse/dense rows through the same wires
    A2 = BlockMatrix.sparse_random(128, 128, 0.04, seed=63)
    if A2.is_local(0, 0):
        A2.set_block(0, 0, torch.randn(A2.block_shape(0, 0)))
    B2 = BlockMatrix.randn(128, 64, seed=64)
    out2 = BlockMatrix(128, 64, A2.block_size, GRID, A2.dtype, A2.device)
    summa_matmul(A2, B2, out2)
    assert torch.allclose(out2.to_dense(),
                          A2.to_dense() @ B2.to_dense(), atol=1e-3)


def _body_fused_transpose_2x2(s, tmpdir):
    """(A op B).T fused path with real 2D routing (both sub-group
    directions non-trivial)."""
    A = s.randn(128, 96, seed=71)
    B = s.randn(128, 96, seed=72)
    out = (A * B).T.compute()
    assert torch.allclose(out.to_tensor(),
                          (A.to_tensor() * B.to_tensor()).t(), atol=1e-5)
    S = s.block_sparse_random(128, 96, 0.4, seed=73)
    out2 = (S + A).T.compute()
    assert torch.allclose(out2.to_tensor(),
                          (S.to_tensor() + A.to_tensor()).t(), atol=1e-5)


@pytest.mark.parametrize("body", [
    "_body_summa_2x2", "_body_chain_2x2", "_body_sparse_2x2",
    "_body_agg_2x2", "_body_ols_2x2", "_body_both_sparse_2x2",
    "_body_fused_transpose_2x2",
])
def test_distributed_world4(body, tmp_path):
    _run(body, tmp_path)


def _body_ols_2x2(s, tmpdir):
    """Config-5 shape class on the 2x2 grid: chain DP + CSE + schemes +
    gather-solve end-to-end."""
    X = s.randn(256, 24, seed=51)
    y = s.randn(256, 1, seed=52)
    w = ((X.T @ X).inverse() @ (X.T @ y)).compute()
    import torch as _t
    xx, yy = X.to_tensor().double(), y.to_tensor().double()
    ref = _t.linalg.lstsq(xx, yy).solution
    assert _t.allclose(w.to_tensor().double(), ref, atol=1e-2)
