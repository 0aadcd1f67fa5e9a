"""OLS pipeline example: w = (X^T X)^-1 X^T y   (BASELINE config 5).

Exercises the optimizer end-to-end: CSE recognizes the shared X^T, chain
ordering computes the small (k x k) and (k x 1) products before the
inverse, the solve runs in fp64 on the gathered k x k Gram matrix
(SURVEY.md §3.5).

Run:  python examples/ols_regression.py [--rows 1000000 --cols 4096]
Distributed: python -m torch.distributed.run --nproc-per-node 8 \
    --master-addr 127.0.0.1 examples/ols_regression.py
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa

import matrel_amd as ma  # noqa


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=0)
    ap.add_argument("--cols", type=int, default=0)
    args = ap.parse_args()
    gpu = torch.cuda.is_available()
    rows = args.rows or (1_000_000 if gpu else 4096)
    cols = args.cols or (4096 if gpu else 128)
    ma.set_config(ma.MatrelConfig(block_size=16384 if gpu else 512))
    s = ma.MatrelSession()

    X = s.randn(rows, cols, seed=1)
    y = s.randn(rows, 1, seed=2)

    w_expr = (X.T @ X).inverse() @ (X.T @ y)
    if s.rank == 0:
        print(w_expr.explain())
    t0 = time.time()
    w = w_expr.compute()
    if s.rank == 0:
        wt = w.to_tensor()
        print(f"solved {rows}x{cols} OLS in {time.time()-t0:.2f}s; "
              f"|w| = {wt.norm().item():.4f}")
        print("metrics:", {k: round(v, 4) if isinstance(v, float) else v
                           for k, v in s.metrics_report().items()
                           if k != "ops"})
    s.shutdown()


if __name__ == "__main__":
    main()
