"""Race-screen + A/B for the counted-vmcnt 2-phase MX-fp8 GEMM (variant 2)
vs the shipped simple k128 structure. Same two-lane discipline as
scripts/deepring_ab.py: multi-run correctness at several sizes against the
shipped kernel AND the bf16 dequant model, then within-process perf A/B.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext

ext = require_ext()


def mx(a):
    a8, sc = ext.mx8_convert(a)
    return a8, sc


def gemm(v, a8, asc, bt8, btsc, n):
    ext.set_mx8_variant(v)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
    return c


def screen(n, runs):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a8, asc = mx(a)
    bt8, btsc = mx(bt)
    ref = gemm(0, a8, asc, bt8, btsc, n)
    ok = True
    for r in range(runs):
        c = gemm(2, a8, asc, bt8, btsc, n)
        err = (c - ref).abs().max().item()
        if err > 0:
            # both kernels consume identical fp8+scale inputs with fp32
            # accum in the same order per C element - must be bit-equal
            print(f"MISMATCH n={n} run={r}: maxdiff={err}")
            ok = False
    print(f"n={n}: screen {'OK' if ok else 'FAIL'}")
    del a, bt, a8, asc, bt8, btsc, ref
    return ok


def perf(n, it, reps):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a8, asc = mx(a)
    bt8, btsc = mx(bt)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    res = {0: [], 2: []}
    for rep in range(reps):
        for v in (0, 2):
            ext.set_mx8_variant(v)
            for _ in range(2):
                ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(it):
                ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / it
            res[v].append(2 * n ** 3 / dt / 1e12)
    for v in (0, 2):
        xs = res[v]
        print(f"n={n} v{v}: " + " ".join(f"{x:.0f}" for x in xs) +
              f"  best={max(xs):.0f} TF")
    del a, bt, a8, asc, bt8, btsc, c


ok = True
for n, runs in ((256, 4), (512, 4), (4096, 3)):
    ok = screen(n, runs) and ok
if not ok:
    print("SCREEN FAILED - not running perf")
    sys.exit(1)
perf(8192, 8, 3)
perf(16384, 3, 2)
ext.set_mx8_variant(0)
print("done")
