"""Triangle counting on an element-sparse undirected graph.

Reference workload family (SURVEY.md §2.1 "Examples / workloads" +
MatRel's joint LA+relational processing): triangles = trace(A^3) / 6 for
a symmetric 0/1 adjacency matrix A. Exercises, end to end:

  * element-sparse CSR block storage (`sparse_random`-style adjacency),
  * the sparse x sparse CSR x CSR block product (csrc/spgemm.hip) for
    A @ A, with the ~1% density crossover routing,
  * matrix-chain planning for A @ A @ A,
  * the trace aggregate push-down (trace(B @ A) rewritten so the final
    n x n product is never materialized),
  * a relational cross-check: semi-join/select-based degree filtering.

Validated against a dense torch reference at small n.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa

import matrel_amd as ma  # noqa


def random_adjacency(n: int, p: float, seed: int = 7) -> torch.Tensor:
    g = torch.Generator().manual_seed(seed)
    upper = (torch.rand(n, n, generator=g) < p).triu(1)
    sym = upper | upper.t()
    return sym.float()


def main():
    gpu = torch.cuda.is_available()
    n = 16384 if gpu else 1024
    p = 0.002 if gpu else 0.01
    ma.set_config(ma.MatrelConfig(block_size=4096 if gpu else 256))
    s = ma.MatrelSession()

    adj = random_adjacency(n, p)
    A = s.from_dense(adj)  # auto-sparsifies blocks below the threshold

    t0 = time.perf_counter()
    # trace(A @ A @ A) / 6: chain DP orders the product; the outermost
    # trace is pushed down so the final product never materializes.
    tri = (A @ A @ A).trace().item() / 6.0
    dt = time.perf_counter() - t0
    print(f"n={n} p={p}: {tri:.0f} triangles  ({dt*1000:.1f} ms)")

    # relational follow-up: how many edges touch high-degree vertices?
    deg = A.row_sums()  # n x 1 vector of degrees
    dmax = deg.max().item()
    hot = deg.select_values(0.5 * dmax, float("inf")).count().item()
    print(f"max degree {dmax:.0f}; vertices above half-max: {hot:.0f}")

    if n <= 4096:
        ref = torch.linalg.matrix_power(adj, 3).diagonal().sum() / 6
        err = abs(tri - ref.item())
        assert err < max(1.0, 1e-3 * abs(ref.item())), (tri, ref.item())
        print(f"matches dense reference ({ref.item():.0f})")


if __name__ == "__main__":
    main()
