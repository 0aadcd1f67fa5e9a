"""Correctness + perf for the 256^2-tile exact-f32 GEMM vs the 128^2 one
and torch fp64 reference."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext

ext = require_ext()


def run(n, it=4, v=0):
    ext.set_f32_variant(v)
    a = torch.randn(n, n, device="cuda")
    bt = torch.randn(n, n, device="cuda")
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    ext.gemm_bt_acc(c, a, bt)
    ref = (a.double() @ bt.double().t())
    rel = ((c.double() - ref).abs().max() / ref.abs().max()).item()
    for _ in range(2):
        ext.gemm_bt_acc(c, a, bt)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        ext.gemm_bt_acc(c, a, bt)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / it
    print(f"n={n} v{v}: {2*n**3/dt/1e12:.1f} TF  relerr={rel:.2e}")
    del a, bt, c, ref


for n in (512, 1024, 4096, 8192, 16384):
    for v in (1, 0):
        run(n, it=6 if n <= 8192 else 3, v=v)
ext.set_f32_variant(0)
print("done")
