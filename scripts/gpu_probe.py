"""Quick GPU perf probe: GEMM TFLOPS, transpose/elementwise bandwidth.
Run on the MI355X box: python scripts/gpu_probe.py"""

import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext  # noqa


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ext = require_ext()
    # A/B the bf16 GEMM variants (within-process interleaved)
    for n in (4096, 8192):
        a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
        bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
        c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
        for v, name in ((1, "simple256"), (4, "phased-nb"), (6, "phased-nb-sprio"), (7, "phased-mb-sprio"), (3, "tile128")):
            ext.set_gemm_variant(v)
            dt = timeit(lambda: ext.gemm_bt_acc(c, a, bt))
            print(f"gemm {name} {n}^3: {2*n**3/dt/1e12:.0f} TFLOPS")
        # correctness cross-check phased vs simple
        ext.set_gemm_variant(1)
        c1 = torch.zeros(n, n, dtype=torch.float32, device="cuda")
        ext.gemm_bt_acc(c1, a, bt)
        for v in (2, 4, 5, 6):
            ext.set_gemm_variant(v)
            c2 = torch.zeros(n, n, dtype=torch.float32, device="cuda")
            ext.gemm_bt_acc(c2, a, bt)
            err = (c1 - c2).abs().max().item()
            print(f"  variant {v} vs simple max|diff| = {err}")
        del a, bt, c, c1, c2
    ext.set_gemm_variant(0)
    # MX-fp8 scaled-MFMA GEMM
    for n in (8192, 16384):
        a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
        bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
        a8, asc = ext.mx8_convert(a)
        bt8, btsc = ext.mx8_convert(bt)
        del a, bt
        c = torch.empty(n, n, dtype=torch.float32, device="cuda")
        dt = timeit(lambda: ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True),
                    iters=6, warmup=2)
        print(f"gemm mx8 {n}^3: {2*n**3/dt/1e12:.0f} TFLOPS")
        del a8, asc, bt8, btsc, c
    for n in (2048, 4096, 8192):
        a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
        bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
        c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
        dt = timeit(lambda: ext.gemm_bt_acc(c, a, bt))
        tf = 2 * n**3 / dt / 1e12
        print(f"gemm_bt bf16 {n}^3: {dt*1e3:.2f} ms  {tf:.0f} TFLOPS")
    # f32 gemm
    n = 4096
    af = torch.randn(n, n, device="cuda")
    btf = torch.randn(n, n, device="cuda")
    cf = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    dt = timeit(lambda: ext.gemm_bt_acc(cf, af, btf))
    print(f"gemm_bt f32 {n}^3: {dt*1e3:.2f} ms  {2*n**3/dt/1e12:.0f} TFLOPS")
    # transpose bf16
    n = 16384
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    dt = timeit(lambda: ext.transpose(a))
    gbs = 2 * n * n * 2 / dt / 1e9
    print(f"transpose bf16 {n}^2: {dt*1e3:.2f} ms  {gbs:.0f} GB/s")
    # elementwise add bf16
    b = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    dt = timeit(lambda: ext.ewise(0, a, b))
    gbs = 3 * n * n * 2 / dt / 1e9
    print(f"ewise add bf16 {n}^2: {dt*1e3:.2f} ms  {gbs:.0f} GB/s")
    # row reduce
    dt = timeit(lambda: ext.reduce2d(a, 0, 1))
    print(f"rowsum bf16 {n}^2: {dt*1e3:.2f} ms  {n*n*2/dt/1e9:.0f} GB/s")
    dt = timeit(lambda: ext.reduce2d(a, 0, 0))
    print(f"colsum bf16 {n}^2: {dt*1e3:.2f} ms  {n*n*2/dt/1e9:.0f} GB/s")
    dt = timeit(lambda: ext.reduce2d(a, 0, -1))
    print(f"fullsum bf16 {n}^2: {dt*1e3:.2f} ms  {n*n*2/dt/1e9:.0f} GB/s")


if __name__ == "__main__":
    main()
