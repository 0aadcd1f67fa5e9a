"""Distributed random-expression fuzz (gloo, world 2): the distributed
executor (SUMMA, repartition, scheme assignment) must match single-process
torch on randomly generated trees."""

import os
import random
import sys

import pytest
import torch
import torch.multiprocessing as mp

N = 64


def _gen_spec(rng, depth):
    if depth == 0 or rng.random() < 0.25:
        return ("leaf", rng.randrange(4))
    op = rng.choice(["add", "sub", "mul", "mm", "t", "scale", "sel",
                     "min", "max"])
    if op in ("add", "sub", "mul", "mm", "min", "max"):
        return (op, _gen_spec(rng, depth - 1), _gen_spec(rng, depth - 1))
    if op == "t":
        return (op, _gen_spec(rng, depth - 1))
    if op == "scale":
        return (op, _gen_spec(rng, depth - 1),
                round(rng.uniform(-2, 2), 3))
    return ("sel", _gen_spec(rng, depth - 1), rng.randrange(0, N // 2),
            rng.randrange(N // 2, N))


def _build(spec, leaves):
    kind = spec[0]
    if kind == "leaf":
        m, t = leaves[spec[1]]
        return m, t.clone()
    if kind in ("add", "sub", "mul", "mm", "min", "max"):
        am, at = _build(spec[1], leaves)
        bm, bt = _build(spec[2], leaves)
        if kind == "add":
            return am + bm, at + bt
        if kind == "sub":
            return am - bm, at - bt
        if kind == "mul":
            return am * bm, at * bt
        if kind == "min":
            return am.elem_min(bm), torch.minimum(at, bt)
        if kind == "max":
            return am.elem_max(bm), torch.maximum(at, bt)
        return (am @ bm) * (1.0 / N), (at @ bt) / N
    if kind == "t":
        am, at = _build(spec[1], leaves)
        return am.T, at.t()
    if kind == "scale":
        am, at = _build(spec[1], leaves)
        return am * spec[2], at * spec[2]
    lo, hi = spec[2], spec[3]
    am, at = _build(spec[1], leaves)
    ref = torch.zeros_like(at)
    ref[lo:hi] = at[lo:hi]
    return am.select_rows(lo, hi), ref


def _worker(rank, world, port, iters=25):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import matrel_amd as ma
    ma.set_config(ma.MatrelConfig(block_size=24))
    s = ma.MatrelSession()
    leaves = [(s.randn(N, N, seed=k), None) for k in (1, 2, 3)]
    # element-sparse (CSR-stored) leaf: random trees route it through
    # ewise densify, SUMMA CSR-wire panels, select masks, transpose
    leaves.append((s.sparse_random(N, N, 0.02, seed=4), None))
    leaves = [(m, m.to_tensor()) for m, _ in leaves]
    rng = random.Random(int(os.environ.get("MATREL_FUZZ_SEED", "77")) + world)
    depth = int(os.environ.get("MATREL_FUZZ_DEPTH", "4"))
    for i in range(iters):
        spec = _gen_spec(rng, depth)
        m, ref = _build(spec, leaves)
        out = m.compute().to_tensor()
        assert torch.allclose(out, ref, atol=2e-3, rtol=2e-3), \
            f"iter {i}: {spec}"
        # aggregate of the same random tree: exercises the push_agg
        # rewrite family (incl. through matmul/transpose/scalar/ewise)
        got = m.sum().item()
        want = ref.sum().item()
        assert abs(got - want) <= 2e-3 * max(1.0, abs(want)) + 2e-2, \
            f"iter {i} sum: {got} vs {want}: {spec}"
        rs = m.row_sums().to_tensor().reshape(-1)
        assert torch.allclose(rs, ref.sum(dim=1), atol=2e-2, rtol=2e-3), \
            f"iter {i} rowsums: {spec}"
    dist.destroy_process_group()


def test_distributed_fuzz():
    mp.spawn(_worker, args=(2, 29810), nprocs=2, join=True)


def test_distributed_fuzz_world4():
    """Same random trees on the 2x2 grid (both sub-group dimensions
    non-trivial - the driver's 4/8-GPU shape class)."""
    mp.spawn(_worker, args=(4, 29812, 12), nprocs=4, join=True)
