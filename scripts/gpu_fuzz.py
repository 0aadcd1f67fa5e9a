"""GPU randomized soak: random expression trees through the full engine on
cuda:0 (HIP kernels only - no eager fallback), compared against torch GPU
fp32 on the same rounded inputs. Exercises ragged blocks, padding
wrappers, sparse blocks, masks and the agg-pushdown rewrites under
random composition. Usage: python scripts/gpu_fuzz.py [iters]
"""
import os
import random
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import matrel_amd as ma  # noqa
from matrel_amd.ops.backend import require_ext  # noqa

require_ext()
N = int(os.environ.get("GPU_FUZZ_N", "600"))
BS = int(os.environ.get("GPU_FUZZ_BS", "256"))
PREC = os.environ.get("GPU_FUZZ_PRECISION", "bf16")
ma.set_config(ma.MatrelConfig(block_size=BS, gemm_precision=PREC))
s = ma.MatrelSession()
# fp8 accumulates larger relative error through deep trees
ATOL, RTOL = {"bf16": (0.15, 0.05), "mx8": (1.5, 0.2),
              "mx6": (1.5, 0.2), "mx4": (6.0, 0.75)}[PREC]

leaves = []
for seed in (1, 2, 3):
    m = s.randn(N, N, seed=seed)
    leaves.append((m, m.to_tensor().float()))
sp = s.block_sparse_random(N, N, 0.5, seed=4)
leaves.append((sp, sp.to_tensor().float()))

rng = random.Random(int(os.environ.get("GPU_FUZZ_SEED", "777")))


def gen(depth):
    if depth == 0 or rng.random() < 0.3:
        return ("leaf", rng.randrange(4))
    op = rng.choice(["add", "sub", "mul", "min", "max", "mm", "t",
                     "scale", "shift", "sel"])
    if op in ("add", "sub", "mul", "min", "max", "mm"):
        return (op, gen(depth - 1), gen(depth - 1))
    if op == "t":
        return (op, gen(depth - 1))
    if op in ("scale", "shift"):
        return (op, gen(depth - 1), round(rng.uniform(-2, 2), 3))
    return ("sel", gen(depth - 1), rng.randrange(0, N // 2),
            rng.randrange(N // 2, N))


def build(spec):
    kind = spec[0]
    if kind == "leaf":
        m, t = leaves[spec[1]]
        return m, t.clone()
    if kind in ("add", "sub", "mul", "min", "max", "mm"):
        am, at = build(spec[1])
        bm, bt = build(spec[2])
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
        am, at = build(spec[1])
        return am.T, at.t()
    if kind == "scale":
        am, at = build(spec[1])
        return am * spec[2], at * spec[2]
    if kind == "shift":
        am, at = build(spec[1])
        return am + spec[2], at + spec[2]
    lo, hi = spec[2], spec[3]
    am, at = build(spec[1])
    ref = torch.zeros_like(at)
    ref[lo:hi] = at[lo:hi]
    return am.select_rows(lo, hi), ref


iters = int(sys.argv[1]) if len(sys.argv) > 1 else 150
bad = 0
for i in range(iters):
    spec = gen(5)
    m, ref = build(spec)
    out = m.compute().to_tensor().float()
    # bf16 engine vs fp32 torch: tolerance scales with tree depth
    if not torch.allclose(out, ref, atol=ATOL, rtol=RTOL):
        err = (out - ref).abs().max().item()
        print(f"MISMATCH iter {i}: maxerr {err:.4f} spec={spec}")
        bad += 1
        if bad > 3:
            break
    if i % 25 == 0:
        sv = m.sum().item()
        rv = ref.sum().item()
        tol = 0.02 if PREC == "bf16" else 0.08
        assert abs(sv - rv) <= tol * max(1.0, abs(rv)) + 8.0, (i, sv, rv)
print(f"gpu fuzz [{PREC}] done: {iters} trees, mismatches: {bad}")
sys.exit(1 if bad else 0)
