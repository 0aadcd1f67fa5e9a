"""Race-screen + A/B for the DEEPRING phased GEMM (variants 8/9) vs the
shipped variant 6.

Sync-structure edits are NEW templates (cdna_hip_programming.md two-lane
discipline): multi-run correctness screen at 256/512/4096 against the
torch matmul reference AND variant 6, then within-process perf A/B.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext

ext = require_ext()
ext.set_supertile(4, 4)


def gemm(v, a, bt, n):
    ext.set_gemm_variant(v)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    ext.gemm_bt_acc(c, a, bt)
    return c


def screen(n, runs):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    ref = gemm(6, a, bt, n)
    tol = 1e-3 * n ** 0.5
    ok = True
    for v in (12, 13):
        for r in range(runs):
            c = gemm(v, a, bt, n)
            err = (c - ref).abs().max().item()
            if err > tol:
                print(f"RACE/MISMATCH v{v} n={n} run={r}: maxdiff={err}")
                ok = False
    # torch reference sanity (v6 itself, loose bf16 band)
    tref = (a.float() @ bt.float().T)
    rerr = (ref - tref).abs().max().item() / max(1.0, tref.abs().max().item())
    print(f"n={n}: screen {'OK' if ok else 'FAIL'}; v6 vs torch relerr={rerr:.2e}")
    del a, bt, ref, tref
    return ok


def perf(n, it, reps):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    res = {6: [], 12: [], 13: []}
    for rep in range(reps):
        for v in (6, 12, 13):
            ext.set_gemm_variant(v)
            for _ in range(2):
                ext.gemm_bt_acc(c, a, bt)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(it):
                ext.gemm_bt_acc(c, a, bt)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / it
            res[v].append(2 * n ** 3 / dt / 1e12)
    for v in (6, 12, 13):
        xs = res[v]
        print(f"n={n} v{v}: " + " ".join(f"{x:.0f}" for x in xs) +
              f"  best={max(xs):.0f} TF")
    del a, bt, c


ok = True
for n, runs in ((256, 4), (512, 4), (4096, 3)):
    ok = screen(n, runs) and ok
if not ok:
    print("SCREEN FAILED - not running perf")
    sys.exit(1)
perf(8192, 8, 3)
perf(16384, 3, 2)
ext.set_gemm_variant(0)
print("done")
