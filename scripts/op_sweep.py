"""Kernel measurement sweep across shapes (committed into profiles/)."""
import os, sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext

ext = require_ext()

def t(fn, it=6, w=2):
    for _ in range(w):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it

print("# bf16 GEMM (auto variant) across N^3")
for n in (2048, 4096, 8192, 16384, 32768):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    c = torch.empty(n, n, dtype=torch.float32, device="cuda")
    it = 6 if n <= 16384 else 3
    dt = t(lambda: ext.gemm_bt_acc_beta0(c, a, bt), it=it)
    print(f"bf16 {n}^3: {2*n**3/dt/1e12:.0f} TF  ({dt*1e3:.2f} ms)")
    del a, bt, c
print("# MX-fp8 GEMM across N^3")
for n in (4096, 8192, 16384, 32768):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a8, asc = ext.mx8_convert(a); bt8, btsc = ext.mx8_convert(bt)
    del a, bt
    c = torch.empty(n, n, dtype=torch.float32, device="cuda")
    it = 6 if n <= 16384 else 3
    dt = t(lambda: ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True), it=it)
    print(f"mx8  {n}^3: {2*n**3/dt/1e12:.0f} TF  ({dt*1e3:.2f} ms)")
    del a8, asc, bt8, btsc, c
print("# bandwidth ops at 16384^2 bf16")
n = 16384
a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
b = torch.randn(n, n, device="cuda").to(torch.bfloat16)
for name, fn, byts in (
    ("transpose", lambda: ext.transpose(a), 2*n*n*2),
    ("ewise add", lambda: ext.ewise(0, a, b), 3*n*n*2),
    ("scalar mul", lambda: ext.ewise_scalar(1, a, 2.0), 2*n*n*2),
    ("mask value", lambda: ext.mask_value(a, 0.0, 1.0, True), 2*n*n*2),
    ("rowsum", lambda: ext.reduce2d(a, 0, 1), n*n*2),
    ("colsum", lambda: ext.reduce2d(a, 0, 0), n*n*2),
    ("fullsum", lambda: ext.reduce2d(a, 0, -1), n*n*2),
    ("mx8 convert", lambda: ext.mx8_convert(a), 3*n*n),
):
    dt = t(fn)
    print(f"{name}: {byts/dt/1e9:.0f} GB/s  ({dt*1e3:.2f} ms)")
