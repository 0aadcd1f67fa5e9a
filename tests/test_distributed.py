"""Multi-process (gloo, world_size=2) tests of the distributed paths:
SUMMA, broadcast-MM, repartition, distributed transpose, aggregates.

This is the analog of the reference's local[N] Spark integration tests
(SURVEY.md §4): real sub-groups, real collectives, no GPU needed.
The same code paths run over RCCL on the GPU box (one rank per MI355X).
"""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

WORLD = 2


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
    from matrel_amd.exec.runtime import reset_runtime
    ma.set_config(ma.MatrelConfig(block_size=32))
    globals()[fn_name](ma.MatrelSession(), tmpdir)
    dist.destroy_process_group()


def _run(fn_name: str, tmp_path) -> None:
    port = 29600 + abs(hash(fn_name)) % 200
    mp.spawn(_worker, args=(WORLD, port, fn_name, str(tmp_path)),
             nprocs=WORLD, join=True)


# ---- bodies (run inside workers) ----------------------------------------

def _body_summa(s, tmpdir):
    A = s.randn(96, 80, seed=1)
    B = s.randn(80, 64, seed=2)
    C = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)


def _body_bcast_mm(s, tmpdir):
    # B small -> planner picks bcast_b; result row-partitioned
    A = s.randn(128, 64, seed=3)
    B = s.randn(64, 32, seed=4)
    expr = A @ B
    from matrel_amd.plan.optimizer import optimize
    from matrel_amd.plan.physical import plan_physical
    opt = optimize(expr.plan)
    pp = plan_physical(opt)
    algo = pp.get(opt).mm_algo
    assert algo in ("bcast_b", "summa")
    C = expr.compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)


def _body_aggregates(s, tmpdir):
    A = s.randn(96, 64, seed=5)
    a = A.to_tensor()
    assert abs(A.sum().item() - a.sum().item()) < 1e-2
    assert abs(A.max().item() - a.max().item()) < 1e-5
    assert torch.allclose(A.row_sums().to_tensor().reshape(-1),
                          a.sum(dim=1), atol=1e-3)
    assert torch.allclose(A.col_sums().to_tensor().reshape(-1),
                          a.sum(dim=0), atol=1e-3)


def _body_transpose(s, tmpdir):
    A = s.randn(96, 64, seed=6)
    At = A.T.compute()
    assert torch.allclose(At.to_tensor(), A.to_tensor().t(), atol=0)


def _body_ewise_transpose_fused(s, tmpdir):
    """(A op B).T takes the fused single-pass path in the executor,
    including with block-sparse operands (union/intersection presence)
    and ragged edges, over real routing collectives."""
    A = s.randn(96, 64, seed=61)
    B = s.randn(96, 64, seed=62)
    for expr, ref in (((A + B).T, (A.to_tensor() + B.to_tensor()).t()),
                      ((A * B).T, (A.to_tensor() * B.to_tensor()).t()),
                      ((A - B).T, (A.to_tensor() - B.to_tensor()).t())):
        out = expr.compute()
        assert torch.allclose(out.to_tensor(), ref, atol=1e-5)
    S = s.block_sparse_random(96, 64, 0.4, seed=63)
    out = (S + A).T.compute()
    assert torch.allclose(out.to_tensor(),
                          (S.to_tensor() + A.to_tensor()).t(), atol=1e-5)
    out = (S * A).T.compute()
    assert torch.allclose(out.to_tensor(),
                          (S.to_tensor() * A.to_tensor()).t(), atol=1e-5)


def _body_repartition(s, tmpdir):
    from matrel_amd.blocks.schemes import BCAST, COL, ROW
    A = s.randn(96, 64, seed=7)
    bm = A.blocks
    for sch in (ROW, COL, BCAST):
        moved = bm.repartition(sch)
        assert torch.allclose(moved.to_dense(), bm.to_dense(), atol=0)


def _body_repartition_csr_wire(s, tmpdir):
    """Element-sparse repartition moves CSR arrays (nnz-proportional
    bytes) and the receivers KEEP sparse storage (round 2; previously
    every scheme move densified the wire AND the received block)."""
    from matrel_amd.blocks.schemes import BCAST, COL, ROW
    from matrel_amd.exec import metrics
    from matrel_amd.ops import blockops
    metrics.reset()
    A = s.sparse_random(128, 96, 0.03, seed=91)
    bm = A.blocks
    for sch in (ROW, COL, BCAST):
        moved = bm.repartition(sch)
        assert torch.allclose(moved.to_dense(), bm.to_dense(), atol=0)
        # received blocks stay CSR
        for _, t in moved.local_items():
            assert blockops.is_sparse(t), sch
    rep = metrics.report()
    keys = rep["collectives"]
    assert any(k.startswith("repartition") and k.endswith("_csr")
               for k in keys), keys
    assert "repartition_bcast" not in keys and \
        "repartition_send" not in keys, keys
    # dense matrices still use the dense wire
    B = s.randn(96, 64, seed=92)
    moved = B.blocks.repartition(ROW)
    assert torch.allclose(moved.to_dense(), B.blocks.to_dense(), atol=0)


def _body_block_sparse(s, tmpdir):
    A = s.block_sparse_random(128, 128, 0.4, seed=8)
    B = s.randn(128, 64, seed=9)
    C = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)


def _body_sparse_skewed(s, tmpdir):
    """Sparse x sparse with skewed block presence: exercises the
    group-uniform broadcast-participation logic (ka/kb) in SUMMA."""
    for seed in (1, 2, 3, 4):
        A = s.block_sparse_random(160, 160, 0.3, seed=seed)
        B = s.block_sparse_random(160, 96, 0.3, seed=seed + 50)
        C = (A @ B).compute()
        ref = A.to_tensor() @ B.to_tensor()
        assert torch.allclose(C.to_tensor(), ref, atol=1e-3), seed


def _body_elem_sparse_summa(s, tmpdir):
    """Element-sparse (CSR-stored) A through SUMMA: the A panel must go
    over the wire as CSR arrays (nnz-proportional bytes), not densified
    bs^2 payloads, and the result must match the dense reference."""
    from matrel_amd.blocks.matrix import BlockMatrix
    from matrel_amd.blocks.schemes import GRID
    from matrel_amd.exec import metrics
    from matrel_amd.exec.summa import summa_matmul
    A = BlockMatrix.sparse_random(96, 96, 0.02, seed=21)  # CSR blocks
    B = BlockMatrix.randn(96, 64, seed=22)
    out = BlockMatrix(96, 64, A.block_size, GRID, A.dtype, A.device)
    summa_matmul(A, B, out)
    ref = A.to_dense() @ B.to_dense()
    assert torch.allclose(out.to_dense(), ref, atol=1e-3)
    rep = metrics.report()
    assert "summa_a_bcast_csr" in rep["collectives"]
    # nnz-proportional: the CSR wire must move far fewer bytes than the
    # densified panels would (2% dense blocks -> ~10x+ reduction incl.
    # index overhead)
    csr_bytes = rep["collectives"]["summa_a_bcast_csr"]["bytes"]
    n_sparse_blocks = len(A.blocks)
    dense_equiv = 0
    for (bi, bk) in A.blocks:
        h, w = A.block_shape(bi, bk)
        dense_equiv += h * w * A.blocks[(bi, bk)].element_size()
    assert csr_bytes < 0.5 * dense_equiv, (csr_bytes, dense_equiv)
    # mixed dense/sparse rows + ragged edges
    A2 = BlockMatrix.sparse_random(100, 88, 0.03, seed=23)
    d = torch.randn(A2.block_shape(0, 0))
    if A2.is_local(0, 0):
        A2.set_block(0, 0, d)  # one dense block among CSR blocks
    B2 = BlockMatrix.randn(88, 72, seed=24)
    out2 = BlockMatrix(100, 72, A2.block_size, GRID, A2.dtype, A2.device)
    summa_matmul(A2, B2, out2)
    ref2 = A2.to_dense() @ B2.to_dense()
    assert torch.allclose(out2.to_dense(), ref2, atol=1e-3)


def _body_both_sparse_summa(s, tmpdir):
    """BOTH operands element-sparse through SUMMA: each panel direction
    must go over the wire as CSR arrays (nnz-proportional bytes on BOTH
    panels - VERDICT r1 items 2+3), the local products must run the
    sparse x sparse path, and the result must match the dense reference."""
    from matrel_amd.blocks.matrix import BlockMatrix
    from matrel_amd.blocks.schemes import GRID
    from matrel_amd.exec import metrics
    from matrel_amd.exec.summa import summa_matmul
    metrics.reset()
    A = BlockMatrix.sparse_random(96, 96, 0.03, seed=25)
    B = BlockMatrix.sparse_random(96, 64, 0.03, seed=26)
    out = BlockMatrix(96, 64, A.block_size, GRID, A.dtype, A.device)
    summa_matmul(A, B, out)
    ref = A.to_dense() @ B.to_dense()
    assert torch.allclose(out.to_dense(), ref, atol=1e-3)
    rep = metrics.report()
    assert "summa_a_bcast_csr" in rep["collectives"], rep["collectives"]
    assert "summa_b_bcast_csr" in rep["collectives"], rep["collectives"]
    # nnz-proportional on BOTH panels: no dense bs^2 payloads at all
    assert "summa_b_bcast" not in rep["collectives"], rep["collectives"]
    for key, m in (("summa_a_bcast_csr", A), ("summa_b_bcast_csr", B)):
        csr_bytes = rep["collectives"][key]["bytes"]
        dense_equiv = sum(
            m.block_shape(bi, bj)[0] * m.block_shape(bi, bj)[1]
            * m.blocks[(bi, bj)].values().element_size()
            for (bi, bj) in m.blocks)
        assert csr_bytes < 0.5 * dense_equiv, (key, csr_bytes, dense_equiv)
    # mixed: sparse A x dense B and dense A x sparse B still correct
    C = BlockMatrix.randn(96, 64, seed=27)
    out2 = BlockMatrix(96, 64, A.block_size, GRID, A.dtype, A.device)
    summa_matmul(A, C, out2)
    assert torch.allclose(out2.to_dense(), A.to_dense() @ C.to_dense(),
                          atol=1e-3)
    D = BlockMatrix.randn(96, 96, seed=28)
    out3 = BlockMatrix(96, 64, A.block_size, GRID, A.dtype, A.device)
    summa_matmul(D, B, out3)
    assert torch.allclose(out3.to_dense(), D.to_dense() @ B.to_dense(),
                          atol=1e-3)


def _body_relational(s, tmpdir):
    """Distributed relational verbs (SURVEY.md §2.1 joins/aggregates over
    matrix data): row-index join (gram), cross-product join + aggregate,
    value semi-join - all over real collectives."""
    A = s.randn(96, 48, seed=31)
    B = s.randn(96, 48, seed=32)
    at, bt = A.to_tensor(), B.to_tensor()
    g = A.join_rows(B).compute()
    assert torch.allclose(g.to_tensor(), at.t() @ bt, atol=1e-2)
    u = s.randn(48, 1, seed=33)
    v = s.randn(1, 64, seed=34)
    total = u.cross_join(v, "mul").sum().item()
    ref = (u.to_tensor() @ v.to_tensor()).sum().item()
    assert abs(total - ref) < 1e-1
    torch.manual_seed(7)  # same tensor on every rank
    t = torch.randint(0, 5, (96, 96)).float()
    C = s.from_dense(t)
    D = s.from_dense(torch.tensor([[1.0, 3.0]]))
    out = C.semi_join_values(D).compute()
    ct = C.to_tensor()
    ref = torch.where((ct == 1) | (ct == 3), ct, torch.zeros(()))
    assert torch.allclose(out.to_tensor(), ref, atol=0)


def _body_mx8_config(s, tmpdir):
    """gemm_precision='mx8'/'mx4' on a CPU/gloo world: the quantized
    branches require CUDA and must fall back UNIFORMLY on every rank (a
    rank-divergent decision would deadlock the panel broadcasts)."""
    import matrel_amd as ma
    cfg = ma.get_config()
    try:
        for prec in ("mx8", "mx4"):
            cfg.gemm_precision = prec
            A = s.randn(96, 96, seed=61)
            B = s.randn(96, 64, seed=62)
            C = (A @ B).compute()
            ref = A.to_tensor() @ B.to_tensor()
            assert torch.allclose(C.to_tensor(), ref, atol=1e-3), prec
    finally:
        cfg.gemm_precision = "bf16"


def _body_deterministic_serialized(s, tmpdir):
    """overlap_comm=False (deterministic/debug mode): SUMMA issues each
    step's broadcasts only after the previous step's compute - same
    numbers, serialized schedule."""
    import matrel_amd as ma
    cfg = ma.get_config()
    cfg.overlap_comm = False
    cfg.deterministic = True
    try:
        A = s.randn(96, 96, seed=41)
        B = s.randn(96, 64, seed=42)
        C = (A @ B).compute()
        ref = A.to_tensor() @ B.to_tensor()
        assert torch.allclose(C.to_tensor(), ref, atol=1e-3)
    finally:
        cfg.overlap_comm = True
        cfg.deterministic = False


def _body_save_load(s, tmpdir):
    A = s.randn(96, 64, seed=10)
    path = os.path.join(tmpdir, "dist_mat")
    A.save(path)
    B = s.load(path)
    assert torch.allclose(A.to_tensor(), B.to_tensor(), atol=0)


def _body_partition_aware_local(s, tmpdir):
    """Schemes already aligned (A row-partitioned, B replicated) => the
    planner must pick the zero-communication 'local' algorithm."""
    from matrel_amd.blocks.schemes import BCAST, ROW
    from matrel_amd.dsl import Matrix
    from matrel_amd.blocks.matrix import BlockMatrix
    from matrel_amd.plan.optimizer import optimize
    from matrel_amd.plan.physical import plan_physical
    A = Matrix.from_blocks(BlockMatrix.randn(96, 64, seed=21, scheme=ROW))
    B = Matrix.from_blocks(
        BlockMatrix.randn(64, 4096, seed=22, scheme=BCAST))
    expr = A @ B
    opt = optimize(expr.plan)
    pp = plan_physical(opt)
    assert pp.get(opt).mm_algo == "local", pp.get(opt).mm_algo
    C = expr.compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-2)


def _body_maxmin_empty_rank(s, tmpdir):
    """Regression: full max/min on a matrix with
    fewer block rows than grid rows - the rank owning ZERO blocks must
    contribute the -/+inf identity, not 0.0, or an all-negative matrix
    reports max=0."""
    # block_size=32, world=2 -> grid (2,1); a 32x64 matrix has 1 block row,
    # so grid-row-1's rank owns nothing under GRID.
    neg = s.from_dense(-torch.rand(32, 64) - 1.0)  # all in [-2, -1]
    m = neg.max().item()
    assert -2.0 <= m <= -1.0, f"max of all-negative matrix was {m}"
    pos = s.from_dense(torch.rand(32, 64) + 1.0)   # all in [1, 2]
    mn = pos.min().item()
    assert 1.0 <= mn <= 2.0, f"min of all-positive matrix was {mn}"


def _body_ewise_move_smaller(s, tmpdir):
    """Regression: when ewise operands differ in
    scheme, the SMALLER operand must be the one repartitioned (planner
    decision), not unconditionally b -> a's scheme."""
    from matrel_amd.blocks.matrix import BlockMatrix
    from matrel_amd.blocks.schemes import COL, ROW
    from matrel_amd.dsl import Matrix
    from matrel_amd.exec import metrics
    from matrel_amd.plan.optimizer import optimize
    from matrel_amd.plan.physical import plan_physical
    # A block-sparse (small side) in ROW scheme, B dense in COL scheme.
    A = Matrix.from_blocks(
        BlockMatrix.block_sparse_random(128, 128, 0.25, seed=71,
                                        scheme=ROW))
    B = Matrix.from_blocks(BlockMatrix.randn(128, 128, seed=72,
                                             scheme=COL))
    expr = A + B
    opt = optimize(expr.plan)
    pp = plan_physical(opt)
    # plan shape: target scheme is the BIGGER (dense) operand's
    assert pp.get(opt).scheme == COL, pp.get(opt).scheme
    metrics.reset()
    C = expr.compute()
    ref = A.to_tensor() + B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)
    rep = metrics.report()
    moved = sum(v["bytes"] for k, v in rep["collectives"].items()
                if k.startswith("repartition"))
    # only A's ~25% present blocks move; B (dense, 128x128 fp32) stays.
    dense_bytes = 128 * 128 * 4
    assert moved < 0.75 * dense_bytes, (moved, dense_bytes)


def _body_bcast_cost_no_threshold(s, tmpdir):
    """replication candidacy is cost-based, not a
    1-GiB constant. A replicable operand far above 1 GiB must still win
    broadcast-MM when its bytes beat the SUMMA panel estimate (plan
    shape only - no data is materialized)."""
    from matrel_amd.blocks.matrix import BlockMatrix
    from matrel_amd.blocks.schemes import ROW
    from matrel_amd.plan import logical as L
    from matrel_amd.plan.physical import plan_physical
    # A: 2M x 64k row-partitioned (dense hint), B: 64k^2 = 8.6 GB bf16
    # (>> the old 1-GiB threshold, << the 16-GiB memory cap).
    A = L.Leaf(BlockMatrix(2_000_000, 65536, block_size=16384,
                           scheme=ROW), density=1.0)
    B = L.Leaf(BlockMatrix(65536, 65536, block_size=16384), density=1.0)
    mm = L.MatMul(A, B)
    pp = plan_physical(mm)
    assert pp.get(mm).mm_algo == "bcast_b", pp.get(mm).mm_algo


def _body_ols(s, tmpdir):
    X = s.randn(128, 16, seed=11)
    y = s.randn(128, 1, seed=12)
    w = ((X.T @ X).inverse() @ (X.T @ y)).compute()
    xx, yy = X.to_tensor().double(), y.to_tensor().double()
    ref = torch.linalg.lstsq(xx, yy).solution
    assert torch.allclose(w.to_tensor().double(), ref, atol=1e-2)


# ---- pytest entry points -------------------------------------------------

@pytest.mark.parametrize("body", [
    "_body_summa", "_body_bcast_mm", "_body_aggregates", "_body_transpose",
    "_body_ewise_transpose_fused",
    "_body_repartition", "_body_repartition_csr_wire",
    "_body_block_sparse", "_body_sparse_skewed",
    "_body_elem_sparse_summa", "_body_both_sparse_summa",
    "_body_relational",
    "_body_deterministic_serialized", "_body_mx8_config",
    "_body_save_load", "_body_partition_aware_local", "_body_ols",
    "_body_maxmin_empty_rank", "_body_ewise_move_smaller",
    "_body_bcast_cost_no_threshold",
])
def test_distributed(body, tmp_path):
    _run(body, tmp_path)
