"""Sweep supertile dims for the phased GEMM at 8192^3 and 16384^3."""
import os, sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext

ext = require_ext()
ext.set_gemm_variant(6)

def t(fn, it=8, w=3):
    for _ in range(w):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it

for n in (8192, 16384):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    it = 8 if n == 8192 else 3
    for sm, sn in ((0, 0), (2, 16), (4, 8), (8, 4), (16, 2), (4, 4), (8, 8), (2, 8), (4, 16)):
        ext.set_supertile(sm, sn)
        dt = t(lambda: ext.gemm_bt_acc(c, a, bt), it=it)
        print(f"n={n} st=({sm},{sn}): {2*n**3/dt/1e12:.0f} TF")
    del a, bt, c
ext.set_supertile(4, 4)
