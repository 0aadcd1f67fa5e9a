"""Race-screen + A/B for the 2-barrier pipelined-ring GEMM (variants
14 = ring+PF0, 15 = ring without cross-tile prefetch) vs shipped v6.

Structure change: 2 barriers/K-tile instead of 4 and
cluster-granular software pipelining so no MFMA cluster head parks on an
LDS drain - targets the 25.2% SQ_WAIT_ANY rocprof measured on v6.
Protocol identical to scripts/deepring_ab.py: multi-run correctness
screen at several sizes vs v6 AND torch fp32, then within-process perf
A/B on both the plain-acc path and the fused beta0+bf16-out path the
bench actually runs.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext

ext = require_ext()
ext.set_supertile(4, 4)
VARIANTS = (14, 15, 16)


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
    for v in VARIANTS:
        for r in range(runs):
            c = gemm(v, a, bt, n)
            err = (c - ref).abs().max().item()
            if err > tol:
                print(f"RACE/MISMATCH v{v} n={n} run={r}: maxdiff={err}")
                ok = False
    # fused final path (beta0 + bf16 out) under the ring
    ext.set_gemm_variant(14)
    cacc = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    ob = torch.empty(n, n, dtype=torch.bfloat16, device="cuda")
    if ext.gemm_bt_final(cacc, ob, a, bt, True):
        err = (ob.float() - ref).abs().max().item()
        if err > tol + 0.02 * ref.abs().max().item():
            print(f"FINAL MISMATCH n={n}: {err}")
            ok = False
    tref = (a.float() @ bt.float().T)
    rerr = (ref - tref).abs().max().item() / max(1.0, tref.abs().max().item())
    print(f"n={n}: screen {'OK' if ok else 'FAIL'}; v6 vs torch "
          f"relerr={rerr:.2e}", flush=True)
    del a, bt, ref, tref
    return ok


def perf(n, it, reps, vs=(6,) + VARIANTS):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    res = {v: [] for v in vs}
    for rep in range(reps):
        for v in vs:
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
    for v in vs:
        xs = res[v]
        print(f"n={n} v{v}: " + " ".join(f"{x:.0f}" for x in xs) +
              f"  best={max(xs):.0f} TF", flush=True)
    del a, bt, c


def perf_final(n, it, reps):
    """The bench path: fused beta0 + bf16-out."""
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    c = torch.empty(n, n, dtype=torch.float32, device="cuda")
    ob = torch.empty(n, n, dtype=torch.bfloat16, device="cuda")
    res = {}
    for v in (6, 16):
        ext.set_gemm_variant(v)
        xs = []
        for rep in range(reps):
            ext.gemm_bt_final(c, ob, a, bt, True)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(it):
                ext.gemm_bt_final(c, ob, a, bt, True)
            torch.cuda.synchronize()
            xs.append(2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12)
        res[v] = xs
        print(f"final n={n} v{v}: " + " ".join(f"{x:.0f}" for x in xs) +
              f"  best={max(xs):.0f} TF", flush=True)
    del a, bt, c, ob


if __name__ == "__main__":
    ok = True
    for n, runs in ((256, 4), (512, 4), (4096, 3)):
        ok = screen(n, runs) and ok
    if not ok:
        print("SCREEN FAILED - not running perf")
        sys.exit(1)
    perf(8192, 8, 3, vs=(6, 2, 7, 16))
    perf(16384, 3, 2, vs=(6, 2, 7, 16))
    perf_final(16384, 3, 2)
    if os.environ.get("RING_AB_BIG") == "1":
        perf_final(65536, 1, 2)
    ext.set_gemm_variant(0)
    print("done")


def zf_diag(n=16384, it=3):
    """Zero-fill vs random for v6 (power-bound diagnostic)."""
    ext.set_gemm_variant(6)
    for name, make in (("randn", lambda: torch.randn(n, n, device="cuda")
                        .to(torch.bfloat16)),
                       ("zeros", lambda: torch.zeros(n, n, device="cuda",
                                                     dtype=torch.bfloat16))):
        a = make()
        bt = make()
        c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
        ext.gemm_bt_acc(c, a, bt)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(it):
            ext.gemm_bt_acc(c, a, bt)
        torch.cuda.synchronize()
        tf = 2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12
        print(f"bf16 v6 zf-diag {name} n={n}: {tf:.0f} TF", flush=True)
        del a, bt, c
