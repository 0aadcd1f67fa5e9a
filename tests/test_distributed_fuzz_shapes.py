"""Shape-directed distributed fuzz (gloo, world 2): random expression
trees over RECTANGULAR, vector-like, and smaller-than-one-block
matrices. The square-matrix fuzz (test_distributed_fuzz.py) cannot
reach shape-dependent paths - ragged trailing blocks on both axes,
k-panels narrower than a block, Nx1 operands through matmul chains -
which is exactly the class where round-2's two ragged-block bugs
(mx-usability cols%128, both-sparse fallback slice) lived.

Generation is type-directed: each node is generated FOR a target shape
(r, c); matmul picks a random inner dimension, transpose flips the
target, elementwise keeps it. Leaves materialize on demand per shape.
"""

import os
import random
import sys

import torch
import torch.multiprocessing as mp

DIMS = [1, 7, 24, 40, 64]  # spans sub-block (bs=24), ragged, multi-block


def _leaf(s, shape, leaves, rng):
    key = (shape, rng.randrange(2))
    if key not in leaves:
        m = s.randn(shape[0], shape[1], seed=hash(key) % 9973)
        leaves[key] = (m, m.to_tensor())
    return leaves[key]


def _gen(s, shape, depth, leaves, rng):
    """Return (Matrix expr, torch ref) of exactly `shape`."""
    if depth == 0 or rng.random() < 0.3:
        m, t = _leaf(s, shape, leaves, rng)
        return m, t.clone()
    op = rng.choice(["mm", "add", "mul", "t", "scale", "min", "max"])
    if op == "mm":
        k = rng.choice(DIMS)
        am, at = _gen(s, (shape[0], k), depth - 1, leaves, rng)
        bm, bt = _gen(s, (k, shape[1]), depth - 1, leaves, rng)
        inv = 1.0 / max(k, 1)
        return (am @ bm) * inv, (at @ bt) * inv
    if op == "t":
        am, at = _gen(s, (shape[1], shape[0]), depth - 1, leaves, rng)
        return am.T, at.t()
    if op == "scale":
        am, at = _gen(s, shape, depth - 1, leaves, rng)
        f = round(rng.uniform(-2, 2), 3)
        return am * f, at * f
    am, at = _gen(s, shape, depth - 1, leaves, rng)
    bm, bt = _gen(s, shape, depth - 1, leaves, rng)
    if op == "add":
        return am + bm, at + bt
    if op == "mul":
        return am * bm, at * bt
    if op == "min":
        return am.elem_min(bm), torch.minimum(at, bt)
    return am.elem_max(bm), torch.maximum(at, bt)


def _worker(rank, world, port, iters=30):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import matrel_amd as ma
    bs = int(os.environ.get("MATREL_FUZZ_BS", "24"))
    ma.set_config(ma.MatrelConfig(block_size=bs))
    s = ma.MatrelSession()
    seed = int(os.environ.get("MATREL_FUZZ_SEED", "101")) + world
    rng = random.Random(seed)
    leaves = {}
    for i in range(iters):
        shape = (rng.choice(DIMS), rng.choice(DIMS))
        spec_rng = random.Random(seed * 1000 + i)
        m, ref = _gen(s, shape, 4, leaves, spec_rng)
        out = m.compute().to_tensor()
        assert out.shape == ref.shape, (i, shape, out.shape, ref.shape)
        assert torch.allclose(out, ref, atol=3e-3, rtol=3e-3), \
            f"iter {i} shape {shape}: max err " \
            f"{(out - ref).abs().max().item()}"
        got = m.sum().item()
        want = ref.sum().item()
        assert abs(got - want) <= 3e-3 * max(1.0, abs(want)) + 3e-2, \
            f"iter {i} sum {got} vs {want}"
    dist.destroy_process_group()


def test_distributed_fuzz_shapes():
    mp.spawn(_worker, args=(2, 29840), nprocs=2, join=True)


def test_distributed_fuzz_shapes_world8():
    """Same shape-directed trees on the 4x2 grid (the driver's 8-GPU
    SCALE shape) - vector/ragged operands across a grid where most
    ranks own zero blocks of the small leaves."""
    mp.spawn(_worker, args=(8, 29845, 12), nprocs=8, join=True)
