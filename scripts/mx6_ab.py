"""MX-fp6 (e2m3) GEMM screen + perf. fp6 runs at the fp4 MFMA rate
(~10 PF dense) with 3 mantissa bits (fp8-class accuracy); the screen
checks the engine kernel against torch fp32 at the e2m3 noise band and
the perf section measures the headline square shapes."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops import blockops
from matrel_amd.ops.backend import require_ext

ext = require_ext()


def screen(n):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a6, asc = blockops.to_mx6(a)
    bt6, btsc = blockops.to_mx6(bt)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    ext.gemm_bt_mx6_acc(c, a6, asc, bt6, btsc, True)
    tref = a.float() @ bt.float().t()
    err = (c - tref).abs()
    sigma = 0.05 * n ** 0.5  # e2m3 product-noise scale for N(0,1)
    mx = err.max().item()
    ok = mx < 6 * sigma
    print(f"mx6 n={n}: maxerr {mx:.2f} (6sig={6*sigma:.2f}) "
          f"{'OK' if ok else 'FAIL'}", flush=True)
    del a, bt, a6, asc, bt6, btsc, c, tref
    return ok


def perf(n, it=3, reps=3):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a6, asc = blockops.to_mx6(a)
    bt6, btsc = blockops.to_mx6(bt)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    xs = []
    for _ in range(reps):
        ext.gemm_bt_mx6_acc(c, a6, asc, bt6, btsc, True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(it):
            ext.gemm_bt_mx6_acc(c, a6, asc, bt6, btsc, True)
        torch.cuda.synchronize()
        xs.append(2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12)
    print(f"mx6 n={n}: " + " ".join(f"{x:.0f}" for x in xs) +
          f"  best={max(xs):.0f} TF", flush=True)
    del a, bt, a6, asc, bt6, btsc, c


def conv_perf(n, it=5):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    blockops.to_mx6(a)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        blockops.to_mx6(a)
    torch.cuda.synchronize()
    gbs = (n * n * (2 + 0.75 + 1 / 32)) / ((time.perf_counter() - t0) / it) / 1e9
    print(f"mx6_convert n={n}: {gbs:.0f} GB/s", flush=True)
    del a


def st_sweep(n, it=3):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a6, asc = blockops.to_mx6(a)
    bt6, btsc = blockops.to_mx6(bt)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    for sm, sn in ((4, 4), (8, 4), (4, 8), (8, 8), (2, 8), (16, 4)):
        ext.set_supertile(sm, sn)
        ext.gemm_bt_mx6_acc(c, a6, asc, bt6, btsc, True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(it):
            ext.gemm_bt_mx6_acc(c, a6, asc, bt6, btsc, True)
        torch.cuda.synchronize()
        tf = 2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12
        print(f"mx6 st({sm},{sn}) n={n}: {tf:.0f} TF", flush=True)
    ext.set_supertile(4, 4)
    del a, bt, a6, asc, bt6, btsc, c


if __name__ == "__main__":
    ok = True
    for n in (256, 512, 4096):
        ok = screen(n) and ok
    if not ok:
        print("SCREEN FAILED")
        sys.exit(1)
    conv_perf(16384)
    perf(8192, it=6)
    perf(16384, it=3)
    perf(32768, it=3)
    if os.environ.get("MX6_ST_SWEEP") == "1":
        st_sweep(16384)
        st_sweep(32768)
    print("done")