"""Checkpointed power iteration: the failure-handling subsystem in use.

The reference inherits lineage-based recompute from Spark; this engine's
design is fail-fast + restart-from-checkpoint. The loop
below saves its state matrix every `every` steps through the on-disk
block format; re-running the script resumes from the last complete
checkpoint instead of step 0 (kill it mid-run to see).

Run: python examples/checkpointed_power_iteration.py [ckpt_dir]
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa

import matrel_amd as ma  # noqa
from matrel_amd.exec.faults import CheckpointedLoop  # noqa

gpu = torch.cuda.is_available()
ma.set_config(ma.MatrelConfig(block_size=4096 if gpu else 128))
s = ma.MatrelSession()

n = 16384 if gpu else 512
A = s.block_sparse_random(n, n, 0.3, seed=7)
norm = A.row_sums()  # no materialized product thanks to agg pushdown

ckpt = sys.argv[1] if len(sys.argv) > 1 else "/tmp/matrel_power_ckpt"
loop = CheckpointedLoop(ckpt, every=5)


def step(i, state):
    if "v" not in state:
        v = s.ones(n, 1).compute()
    else:
        v = ma.Matrix.from_blocks(state["v"])
    w = (A @ v).compute()
    scale = 1.0 / max(abs(w.max().item()), 1e-9)
    v2 = (w * scale).compute()
    if s.rank == 0 and i % 5 == 0:
        print(f"step {i}: |w|_max scale {scale:.3e}")
    return {"v": v2.blocks}


final = loop.run(step, n_steps=20)
v = ma.Matrix.from_blocks(final["v"])
print(f"done: |v|_fro = {v.norm_fro().item():.4f} "
      f"(resumed runs skip completed steps)")
