"""Relational analytics over matrix data - MatRel's signature workload
shape (selection, aggregation, joins optimized JOINTLY with
the linear algebra).

Pipeline: a user x item interaction matrix ->
  1. select the active-user row range and positive interactions (select
     on index + value; whole excluded blocks are pruned at plan time),
  2. co-occurrence via a row-index join with multiply-sum combiner
     (lowered to A^T A on the MFMA path; CSE shares the select subtree),
  3. per-item aggregate + top-band mask via value selection.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa

import matrel_amd as ma  # noqa


def main():
    gpu = torch.cuda.is_available()
    users = 262144 if gpu else 2048
    items = 16384 if gpu else 512
    ma.set_config(ma.MatrelConfig(block_size=8192 if gpu else 256))
    s = ma.MatrelSession()

    R = s.rand(users, items, seed=7)  # interaction strengths in [0,1)
    t0 = time.time()

    # 1. relational select: active users (first half), strong interactions
    active = R.select(rows=(0, users // 2), values=(0.5, 1.0))

    # 2. item co-occurrence: join on the user index, multiply-sum combine
    cooc = active.join_rows(active)  # == active^T @ active

    # 3. aggregate: total co-occurrence mass per item + overall stats
    per_item = cooc.row_sums().compute()
    total = cooc.sum().item()
    hot = cooc.select_values(float(0.8 * per_item.to_tensor().max()
                                   / items), float("inf")).count().item()

    if s.rank == 0:
        print(f"co-occurrence {items}x{items} from {users}x{items} "
              f"interactions in {time.time()-t0:.2f}s")
        print(f"  total mass {total:.3e}; hot entries {hot:.0f}")
        print("  plan:")
        print(cooc.sum().explain())
    s.shutdown()


if __name__ == "__main__":
    main()
