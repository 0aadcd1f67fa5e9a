"""PageRank-style iteration on a block-sparse adjacency matrix.

Reference workload family (SURVEY.md §2.1 "Examples / workloads":
PageRank-like iterations). Demonstrates block-sparse SpMM through the
SUMMA/local matmul path (absent blocks skipped), scalar ops and
aggregations.

r_{t+1} = d * (A_norm @ r_t) + (1 - d) / n
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa

import matrel_amd as ma  # noqa


def main():
    gpu = torch.cuda.is_available()
    n = 262144 if gpu else 2048
    block_density = 0.05
    ma.set_config(ma.MatrelConfig(block_size=8192 if gpu else 256))
    s = ma.MatrelSession()

    A = s.block_sparse_random(n, n, block_density, seed=1)
    # normalize by the largest |column sum| so the iteration is stable
    # (a true PageRank would divide each column by its sum; we fold the
    # normalization into one scalar to keep the demo self-contained)
    colsum = A.col_sums().compute()
    scale = float(colsum.to_tensor().abs().max().item()) or 1.0
    An = A * (1.0 / scale)

    r = s.from_dense(torch.full((n, 1), 1.0 / n))
    d = 0.85
    t0 = time.time()
    for it in range(5):
        r = ((An @ r) * d + (1 - d) / n).compute()
        total = r.sum().item()
        if s.rank == 0:
            print(f"iter {it}: sum(r) = {total:.6f}")
    if s.rank == 0:
        print(f"5 iterations on {n}x{n} @ {block_density:.0%} block density "
              f"in {time.time()-t0:.2f}s")
    s.shutdown()


if __name__ == "__main__":
    main()
