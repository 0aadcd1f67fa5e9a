"""Config 4 at the TRUE BASELINE size on ONE GPU: 262144^2 sparse A at
1% block density x a 262144 x 16384 dense strip (the per-GPU slice of
the 8-GPU config). Sparse A ~1.4 GB + strip 8 GB + out 8 GB fits easily;
this turns the r1 'scale-down only' note into a full-size datapoint."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import matrel_amd as ma  # noqa

n = 262144
strip = 16384
bsz = 2048
density = 0.01
ma.set_config(ma.MatrelConfig(block_size=bsz))
s = ma.MatrelSession()
A = s.block_sparse_random(n, n, density, seed=1)
B = s.randn(n, strip, seed=2)
present = int(A.blocks.presence().sum().item())
C = (A @ B).compute()  # warmup
torch.cuda.synchronize()
t0 = time.perf_counter()
iters = 3
for _ in range(iters):
    C = (A @ B).compute()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
flops = 2.0 * present * bsz * bsz * strip
print({"config": "4-blocksparse-spmm-fullN", "n": n, "strip": strip,
       "block": bsz, "block_density": density, "present_blocks": present,
       "sec": round(dt, 4),
       "effective_tflops": round(flops / dt / 1e12, 1)})
