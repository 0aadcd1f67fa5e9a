"""Benchmark the BASELINE.json configs that fit the available GPUs.

  config 2: 65536^2 dense A@B bf16 (also what bench.py measures at 1 GPU)
  config 4: block-sparse x dense SpMM at 1% block density
  config 5: OLS (X^T X)^-1 X^T y with X = 1M x 4096

Writes a JSON line per config. Config 3 (256k, 8 GPUs) runs via
torchrun + bench.py --gpus 8; config 1 (CPU) via bench.py on a CPU host.
"""

import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import matrel_amd as ma  # noqa


def timeit(fn, iters=2, warmup=1):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def config4_spmm():
    n = 131072
    bsz = 2048
    density = 0.01
    ma.set_config(ma.MatrelConfig(block_size=bsz))
    s = ma.MatrelSession()
    A = s.block_sparse_random(n, n, density, seed=1)
    nblocks = len(A.blocks.blocks)
    # dense B: n x 16384 strip (full n x n dense B would dwarf the sparse
    # work; the strip matches SpMM benchmarks' tall-skinny use)
    B = s.randn(n, 16384, seed=2)
    dt = timeit(lambda: (A @ B).compute())
    flops = 2.0 * nblocks * bsz * bsz * 16384
    print(json.dumps({
        "config": "4-blocksparse-spmm", "n": n, "block": bsz,
        "block_density": density, "present_blocks": nblocks,
        "sec": round(dt, 4), "effective_tflops": round(flops / dt / 1e12, 1),
    }))


def config5_ols():
    rows, cols = 1_000_000, 4096
    ma.set_config(ma.MatrelConfig(block_size=16384))
    s = ma.MatrelSession()
    X = s.randn(rows, cols, seed=1)
    y = s.randn(rows, 1, seed=2)
    expr = (X.T @ X).inverse() @ (X.T @ y)
    dt = timeit(lambda: expr.compute(), iters=1, warmup=1)
    flops = 2.0 * rows * cols * cols + 2.0 * rows * cols
    print(json.dumps({
        "config": "5-ols", "rows": rows, "cols": cols, "sec": round(dt, 3),
        "tflops": round(flops / dt / 1e12, 1),
    }))


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("all", "4"):
        config4_spmm()
    if which in ("all", "5"):
        config5_ols()
