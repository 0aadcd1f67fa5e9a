"""RCCL surface tests on a real MI355X.

The engine's entire collective surface ran only on gloo/CPU in round 1.
These tests execute the ACTUAL RCCL (torch.distributed "nccl" backend)
calls the engine makes, in the exact dtype/op combinations it uses, on
one GPU:

* world-1 nccl: real communicator init + every collective pattern from
  blocks/matrix.py (presence: uint8 all_reduce MAX; nnz_map: int64
  all_reduce SUM), exec/summa.py (bf16 / uint8 stacked-panel broadcasts,
  int32 + bf16 variable-size CSR-wire broadcasts) and exec/executor.py
  (fp32 all_reduce MAX/MIN/SUM with +/-inf identities).
* a full engine pipeline (SUMMA-shaped matmul, aggregates, OLS) running
  under an initialized nccl process group.

A world-1 NCCL collective still goes through RCCL's enqueue/stream
machinery and dtype dispatch, so unsupported-dtype or wrong-stream bugs
surface here rather than in the driver's first 8-GPU run. Multi-rank
behavior (sub-group broadcast ordering, CSR wire sizing) is covered by
the gloo world-2/4/8 tests; the remaining gloo-vs-RCCL risks are listed
in docs/RCCL_CHECKLIST.md.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture()
def nccl_world1():
    import torch.distributed as dist
    from matrel_amd.exec.runtime import reset_runtime
    reset_runtime()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29417")
    os.environ["HSA_ENABLE_IPC_MODE_LEGACY"] = "0"
    dist.init_process_group("nccl", rank=0, world_size=1)
    torch.cuda.set_device(0)
    yield dist
    dist.destroy_process_group()
    reset_runtime()


def test_rccl_collective_dtype_surface(nccl_world1):
    """Every (dtype, op) pair the engine hands to RCCL, on device
    tensors, through a real communicator."""
    dist = nccl_world1
    dev = torch.device("cuda", 0)
    # presence map: uint8 MAX (blocks/matrix.py presence())
    pm = torch.randint(0, 2, (8, 8), dtype=torch.uint8, device=dev)
    want = pm.clone()
    dist.all_reduce(pm, op=dist.ReduceOp.MAX)
    assert torch.equal(pm, want)
    # nnz map: int64 SUM (blocks/matrix.py nnz_map())
    nm = torch.randint(0, 1 << 40, (8, 8), dtype=torch.int64, device=dev)
    want = nm.clone()
    dist.all_reduce(nm, op=dist.ReduceOp.SUM)
    assert torch.equal(nm, want)
    # aggregate identities: fp32 MAX with -inf / MIN with +inf partials
    # (executor.py empty-rank identity fix) - RCCL must propagate inf.
    t = torch.tensor([float("-inf")], device=dev)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    assert t.item() == float("-inf")
    t = torch.tensor([float("inf")], device=dev)
    dist.all_reduce(t, op=dist.ReduceOp.MIN)
    assert t.item() == float("inf")
    # SUMMA panel broadcasts: stacked bf16 panels, async + wait
    panel = torch.randn(4, 256, 256, device=dev).bfloat16()
    want = panel.clone()
    w = dist.broadcast(panel, src=0, async_op=True)
    w.wait()
    assert torch.equal(panel, want)
    # CSR wire: int32 index buffer + bf16 value buffer, variable sizes
    idx = torch.randint(0, 1 << 20, (4097,), dtype=torch.int32, device=dev)
    val = torch.randn(993, device=dev).bfloat16()
    wi = dist.broadcast(idx, src=0, async_op=True)
    wv = dist.broadcast(val, src=0, async_op=True)
    wi.wait()
    wv.wait()
    # MX-fp8 wire: uint8 payload + uint8 scales
    p8 = torch.randint(0, 255, (2, 256, 256), dtype=torch.uint8,
                       device=dev)
    dist.broadcast(p8, src=0)
    # fp32 block broadcast (to_dense gather path)
    blk = torch.randn(128, 128, device=dev)
    dist.broadcast(blk, src=0)
    torch.cuda.synchronize()


def test_engine_pipeline_under_nccl(nccl_world1):
    """Full DSL -> optimizer -> executor pipeline with an initialized
    nccl process group: runtime init (device binding, grid), matmul,
    aggregates, OLS. World 1, so per-matrix collectives short-circuit,
    but the session/runtime nccl code path is the real one."""
    import matrel_amd as ma
    s = ma.MatrelSession(ma.MatrelConfig(block_size=512))
    assert s.runtime.backend == "nccl"
    assert s.runtime.device.type == "cuda"
    A = s.randn(1024, 768, seed=1)
    B = s.randn(768, 512, seed=2)
    C = (A @ B).compute()
    ref = A.to_tensor() @ B.to_tensor()
    err = (C.to_tensor() - ref).abs().max() / ref.abs().max()
    assert err < 1e-2, err.item()
    a = A.to_tensor()
    assert abs(A.max().item() - a.max().item()) < 1e-5
    # OLS pipeline (chain ordering + CSE + solve)
    X = s.randn(2048, 64, seed=3)
    y = s.randn(2048, 1, seed=4)
    w = ((X.T @ X).inverse() @ (X.T @ y)).compute()
    xx = X.to_tensor().double()
    yy = y.to_tensor().double()
    refw = torch.linalg.lstsq(xx, yy).solution
    assert torch.allclose(w.to_tensor().double(), refw, atol=5e-2)
