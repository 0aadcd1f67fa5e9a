"""Isolated measurement of the within-block sparse kernels (VERDICT r1
weak #5 / next-round #6): spmm_csr_acc (CSR x dense) and the new
spgemm_csr_csr (CSR x CSR), across densities, with achieved GB/s against
the kernel's algorithmic traffic and the densify+MFMA alternative.

Traffic model (spmm): the kernel reads, per A-nonzero, the full B row
span it contributes to: nnz * N * elem bytes of B (L1/L2 absorb some),
plus C read+write M*N*8 and the CSR arrays. The densify+MFMA alternative
pays 2*M*K*N flops at the measured dense rate plus a densify pass.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops import blockops
from matrel_amd.ops.backend import require_ext

ext = require_ext()


def bench(fn, warm=2, it=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it


def run(n, d):
    g = torch.Generator().manual_seed(42)
    a = torch.randn(n, n, generator=g) * (torch.rand(n, n, generator=g) < d)
    a_bf = a.to(torch.bfloat16).cuda()
    a_csr = a_bf.to_sparse_csr()
    b = torch.randn(n, n, generator=g).to(torch.bfloat16).cuda()
    bt = b.t().contiguous()
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    nnz = a_csr._nnz()

    t_spmm = bench(lambda: blockops.spmm_acc(c, a_csr, b))
    b_traffic = nnz * n * 2          # B bytes touched (pre-cache)
    c_traffic = n * n * 8            # C read+write
    gbs = (b_traffic + c_traffic) / t_spmm / 1e9
    eff_tf = 2 * nnz * n / t_spmm / 1e12

    # densify + fused MFMA alternative on the same operands
    def densified():
        ad = a_csr.to_dense()
        ext.gemm_bt_acc_beta0(c, ad, bt)
    t_dense = bench(densified)

    # spgemm (both sparse) at same density
    b_sp = (torch.randn(n, n, generator=g)
            * (torch.rand(n, n, generator=g) < d)).to(torch.bfloat16).cuda()
    b_csr = b_sp.to_sparse_csr()
    t_spg = bench(lambda: blockops.spgemm_acc(c, a_csr, b_csr)) \
        if n <= 16384 else float("nan")

    print(f"n={n} d={d:.4f} nnz={nnz}: spmm {t_spmm*1e3:.2f} ms "
          f"({gbs:.0f} GB/s algo-traffic, {eff_tf:.1f} eff TF) | "
          f"densify+MFMA {t_dense*1e3:.2f} ms | spgemm {t_spg*1e3:.2f} ms",
          flush=True)
    del a, a_bf, a_csr, b, bt, c, b_sp, b_csr


if __name__ == "__main__":
    for d in (0.001, 0.005, 0.01, 0.02, 0.05):
        run(8192, d)
    for d in (0.005, 0.02):
        run(16384, d)
    print("done")
