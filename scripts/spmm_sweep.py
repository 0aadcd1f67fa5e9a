"""Config-4 evidence: block-sparse x dense SpMM across block densities
(131072^2 sparse A x 131072x16384 dense strip, block 2048)."""
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import matrel_amd as ma  # noqa


def timeit(fn, iters=5, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


n = 131072
bsz = 2048
for density in (0.005, 0.01, 0.02, 0.05):
    ma.set_config(ma.MatrelConfig(block_size=bsz))
    s = ma.MatrelSession()
    A = s.block_sparse_random(n, n, density, seed=1)
    nblocks = len(A.blocks.blocks)
    B = s.randn(n, 16384, seed=2)
    dt = timeit(lambda: (A @ B).compute())
    flops = 2.0 * nblocks * bsz * bsz * 16384
    print(json.dumps({
        "block_density": density, "present_blocks": nblocks,
        "ms": round(dt * 1e3, 2),
        "effective_tflops": round(flops / dt / 1e12, 1)}))
    del A, B
    torch.cuda.empty_cache()
