"""GPU stress validation: randomized engine expressions at real sizes,
multiple shapes/seeds - catches latent bugs the quick numerics tests miss.
Run on the MI355X box; exits nonzero on any mismatch."""

import os
import random
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import matrel_amd as ma  # noqa
from matrel_amd.exec.runtime import reset_runtime  # noqa


def check(tag, out, ref, tol):
    err = (out - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    rel = err / scale
    status = "ok" if rel < tol else "FAIL"
    print(f"  {tag}: max_abs={err:.3e} rel={rel:.3e} {status}")
    assert rel < tol, tag


def run_block(block_size, n, seed):
    reset_runtime()
    ma.set_config(ma.MatrelConfig(block_size=block_size))
    s = ma.MatrelSession()
    rng = random.Random(seed)
    A = s.randn(n, n, seed=seed)
    B = s.randn(n, n, seed=seed + 1)
    S = s.block_sparse_random(n, n, 0.3, seed=seed + 2)
    E = s.sparse_random(n, n, 0.02, seed=seed + 3)
    a, b, sp, e = (A.to_tensor(), B.to_tensor(), S.to_tensor(),
                   E.to_tensor())
    print(f"block={block_size} n={n} seed={seed}")
    check("matmul", (A @ B).compute().to_tensor(), a @ b, 2e-2)
    check("chain", (A @ B @ A).compute().to_tensor(), a @ b @ a, 5e-2)
    check("gram_T", (A.T @ A).compute().to_tensor(), a.t() @ a, 2e-2)
    check("spmm_block", (S @ B).compute().to_tensor(), sp @ b, 2e-2)
    check("spmm_csr", (E @ B).compute().to_tensor(), e @ b, 2e-2)
    check("mixed", ((A + B) @ (A - B)).compute().to_tensor(),
          (a + b) @ (a - b), 5e-2)
    check("select_mm",
          (A.select_rows(0, n // 2) @ B).compute().to_tensor(),
          torch.cat([a[: n // 2], torch.zeros_like(a[n // 2:])]) @ b, 2e-2)
    lo = float(a.min().item())
    check("transposed_out", (A @ B).T.compute().to_tensor(),
          (a @ b).t(), 2e-2)
    su = (A @ B).sum().item()
    assert abs(su - (a @ b).sum().item()) / (abs(su) + 1) < 2e-2, "agg sum"
    print("  aggregates ok")


def main():
    for bs, n, seed in ((1024, 4096, 11), (2048, 6144, 23),
                        (512, 3072, 37), (4096, 8192, 41)):
        run_block(bs, n, seed)
    print("STRESS PASS")


if __name__ == "__main__":
    main()
