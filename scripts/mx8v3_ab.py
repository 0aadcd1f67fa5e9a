"""Race-screen + A/B for MX-fp8 k128 v3 (scale prefetch) vs the shipped
simple k128. v3 rotates the 6 per-tile scale words
one K-tile ahead so the first mfma_scale of a tile no longer waits on
24 strided L2 byte loads. Bit-exactness screen: same quantized inputs
=> v3 must match variant-0 EXACTLY (identical arithmetic, same order).
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops import blockops
from matrel_amd.ops.backend import require_ext

ext = require_ext()


def run(v, a8, asc, bt8, btsc, n):
    ext.set_mx8_variant(v)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
    return c


def screen(n, runs=3):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a8, asc = blockops.to_mx8(a)
    bt8, btsc = blockops.to_mx8(bt)
    ref = run(0, a8, asc, bt8, btsc, n)
    ok = True
    for v in (3, 4, 5):
        for r in range(runs):
            c = run(v, a8, asc, bt8, btsc, n)
            if not torch.equal(c, ref):
                md = (c - ref).abs().max().item()
                print(f"MISMATCH v{v} n={n} run={r}: {md}")
                ok = False
    # sanity vs fp32 torch (loose fp8 band)
    tref = a.float() @ bt.float().t()
    rerr = (ref - tref).abs().max().item() / tref.abs().max().item()
    print(f"n={n}: screen {'OK' if ok else 'FAIL'} (v0 vs torch "
          f"rel {rerr:.3f})", flush=True)
    del a, bt, a8, asc, bt8, btsc, ref, tref
    return ok


def perf(n, it=3, reps=3):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a8, asc = blockops.to_mx8(a)
    bt8, btsc = blockops.to_mx8(bt)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    for v in (0, 3, 4, 5):
        ext.set_mx8_variant(v)
        xs = []
        for _ in range(reps):
            ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(it):
                ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
            torch.cuda.synchronize()
            xs.append(2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12)
        print(f"mx8 n={n} v{v}: " + " ".join(f"{x:.0f}" for x in xs) +
              f"  best={max(xs):.0f} TF", flush=True)
    del a, bt, a8, asc, bt8, btsc, c


def st_sweep(n, it=3):
    """Supertile sweep for the (possibly HBM-bound) mx8 v3 kernel: at
    256 fp8 flops/byte the kernel sits near the bandwidth knee, so L2
    strip reuse through the supertile shape is a first-order lever."""
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a8, asc = blockops.to_mx8(a)
    bt8, btsc = blockops.to_mx8(bt)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    ext.set_mx8_variant(3)
    for sm, sn in ((4, 4), (8, 4), (4, 8), (8, 8), (2, 8), (16, 4)):
        ext.set_supertile(sm, sn)
        ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(it):
            ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
        torch.cuda.synchronize()
        tf = 2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12
        print(f"mx8 st({sm},{sn}) n={n}: {tf:.0f} TF", flush=True)
    ext.set_supertile(4, 4)
    del a, bt, a8, asc, bt8, btsc, c


if __name__ == "__main__":
    ok = True
    for n in (256, 512, 4096):
        ok = screen(n) and ok
    if not ok:
        print("SCREEN FAILED")
        sys.exit(1)
    perf(8192, it=6)
    perf(16384, it=3)
    if os.environ.get("MX8_ST_SWEEP") == "1":
        st_sweep(16384)
        st_sweep(32768)
    ext.set_mx8_variant(0)
    print("done")


def zf_diag(n=16384, it=3):
    """Zero-fill vs random diagnostic (DVFS give-back test, microarch
    guide "DVFS give-back" item 1): if zero-filled operands run much
    faster at the same kernel, the random-data number is POWER-bound
    (toggling energy), not issue/pipe-bound - structural work cannot
    recover it."""
    ext.set_mx8_variant(3)
    for name, make in (("randn", lambda: torch.randn(n, n, device="cuda")
                        .to(torch.bfloat16)),
                       ("zeros", lambda: torch.zeros(n, n, device="cuda",
                                                     dtype=torch.bfloat16))):
        a = make()
        bt = make()
        a8, asc = blockops.to_mx8(a)
        bt8, btsc = blockops.to_mx8(bt)
        c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
        ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(it):
            ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
        torch.cuda.synchronize()
        tf = 2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12
        print(f"mx8 zf-diag {name} n={n}: {tf:.0f} TF", flush=True)
        del a, bt, a8, asc, bt8, btsc, c
