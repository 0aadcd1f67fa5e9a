"""World-size-8 gloo test: the 4x2 grid is EXACTLY the shape the driver's
8-GPU SCALE run uses (one rank per MI355X over RCCL). One spawn, one
combined body, to keep suite time bounded."""

import os
import sys

import torch
import torch.multiprocessing as mp


def _worker(rank: int, world: int, port: int) -> None:
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      HSA_ENABLE_IPC_MODE_LEGACY="0")
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import matrel_amd as ma
    ma.set_config(ma.MatrelConfig(block_size=32))
    s = ma.MatrelSession()
    assert s.runtime.grid == (4, 2), s.runtime.grid
    # dense SUMMA on the 4x2 grid (bench shape class, ragged)
    A = s.randn(160, 144, seed=81)
    B = s.randn(144, 96, seed=82)
    C = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)
    # element-sparse + transpose + aggregate through the same grid
    E = s.sparse_random(160, 128, 0.03, seed=83)
    D = (E.T @ A).compute()
    refd = E.to_tensor().t() @ A.to_tensor()
    assert torch.allclose(D.to_tensor(), refd, atol=1e-3)
    assert abs((A @ B).sum().item() - ref.sum().item()) < 1e-1
    # SUMMA tunables the driver can sweep without code changes
    # (MATREL_SUMMA_PREFETCH / MATREL_OVERLAP_COMM / MATREL_GRID): same
    # numbers at every prefetch depth, including serialized mode.
    cfg = ma.get_config()
    for depth, overlap in ((2, True), (3, True), (1, False)):
        cfg.summa_prefetch = depth
        cfg.overlap_comm = overlap
        C2 = (A @ B).compute()
        assert torch.allclose(C2.to_tensor(), ref, atol=1e-3), \
            (depth, overlap)
    cfg.summa_prefetch = 1
    cfg.overlap_comm = True
    dist.destroy_process_group()


def _worker_grid24(rank: int, world: int, port: int) -> None:
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      HSA_ENABLE_IPC_MODE_LEGACY="0",
                      MATREL_GRID="2x4")
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import matrel_amd as ma
    cfg = ma.MatrelConfig.from_env()
    cfg.block_size = 32
    ma.set_config(cfg)
    s = ma.MatrelSession()
    assert s.runtime.grid == (2, 4), s.runtime.grid
    A = s.randn(160, 144, seed=85)
    B = s.randn(144, 96, seed=86)
    C = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)
    dist.destroy_process_group()


def _worker_stress(rank: int, world: int, port: int) -> None:
    """SCALE-run rehearsal: maximum collective-ordering pressure on the
    exact 4x2 grid - many k-steps, mixed block-/element-sparse operands
    (skipped ka/kb steps + variable-size CSR wires in BOTH directions),
    prefetch depth 3. A deadlock here is a deadlock in the driver's
    8-GPU run; a pass pins the enqueue-order reasoning of
    docs/RCCL_CHECKLIST.md item 2."""
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      HSA_ENABLE_IPC_MODE_LEGACY="0")
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import matrel_amd as ma
    cfg = ma.MatrelConfig(block_size=16)
    cfg.summa_prefetch = 3
    ma.set_config(cfg)
    s = ma.MatrelSession()
    assert s.runtime.grid == (4, 2)
    # 20 k-steps, block-sparse A x element-sparse B
    A = s.block_sparse_random(256, 320, 0.25, seed=101)
    B = s.sparse_random(320, 128, 0.04, seed=102)
    C = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    assert torch.allclose(C.to_tensor(), ref, atol=1e-3)
    # element-sparse x block-sparse the other way, chained with an ewise
    D = s.sparse_random(128, 256, 0.03, seed=103)
    E = s.block_sparse_random(320, 96, 0.3, seed=104)
    F = ((D @ A @ E) * 2.0).compute()
    reff = (D.to_tensor() @ A.to_tensor() @ E.to_tensor()) * 2.0
    assert torch.allclose(F.to_tensor(), reff, atol=1e-2)
    dist.destroy_process_group()


def test_distributed_world8():
    mp.spawn(_worker, args=(8, 29931), nprocs=8, join=True)


def test_distributed_world8_scale_rehearsal():
    mp.spawn(_worker_stress, args=(8, 29951), nprocs=8, join=True)


def test_distributed_world8_grid_2x4():
    """MATREL_GRID=2x4: the transposed grid shape the driver may sweep
    against the default 4x2 on the 8-GPU SCALE run."""
    mp.spawn(_worker_grid24, args=(8, 29941), nprocs=8, join=True)
