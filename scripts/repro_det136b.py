"""Bisect the iter-136 mismatch: evaluate every subtree at (678,584)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import matrel_amd as ma  # noqa
from matrel_amd.exec.runtime import reset_runtime  # noqa

N = 1088
R, C = 678, 584

reset_runtime()
ma.set_config(ma.MatrelConfig(block_size=256))
s = ma.MatrelSession()
leaves = []
for seed in (1, 2, 3):
    m = s.randn(N, N, seed=seed)
    leaves.append((m, m.to_tensor().float()))
sp = s.block_sparse_random(N, N, 0.5, seed=4)
leaves.append((sp, sp.to_tensor().float()))
l1, t1 = leaves[1]
l2, t2 = leaves[2]
l3, t3 = leaves[3]

T1 = l3 + (-1.329)
rT1 = t3 - 1.329
S1 = (l1 * -1.412) - (l2.select_rows(453, 934) - (l3 * l3))
rsel = torch.zeros_like(t2)
rsel[453:934] = t2[453:934]
rS1 = (t1 * -1.412) - (rsel - t3 * t3)
S2 = ((l1 * l1) + (-1.022)) - (l3 - (l1 @ l3) * (1.0 / N))
rS2 = (t1 * t1 - 1.022) - (t3 - (t1 @ t3) / N)
T2 = S1 * S2
rT2 = rS1 * rS2
FULL = T1.elem_max(T2)
rFULL = torch.maximum(rT1, rT2)

for name, expr, ref in (("T1", T1, rT1), ("S1", S1, rS1),
                        ("S2", S2, rS2), ("T2", T2, rT2),
                        ("FULL", FULL, rFULL)):
    out = expr.compute().to_tensor().float()
    err = (out - ref).abs()
    print(f"{name}: maxerr {err.max().item():.4f} "
          f"@({R},{C}) out={out[R, C].item():.4f} "
          f"ref={ref[R, C].item():.4f}", flush=True)

# arbiter: the same S1/T2 computed with TORCH bf16 stepwise - if torch
# agrees with the engine, the mismatch is bf16 cancellation, not logic
b1, b2, b3 = (t1.to(torch.bfloat16), t2.to(torch.bfloat16),
              t3.to(torch.bfloat16))
bsel = torch.zeros_like(b2)
bsel[453:934] = b2[453:934]
bS1 = ((b1 * torch.tensor(-1.412, dtype=torch.bfloat16))
       - (bsel - b3 * b3))
bS2 = ((b1 * b1 + torch.tensor(-1.022).to(torch.bfloat16)).float()
       - (b3.float() - (b1.float() @ b3.float()) / N)).to(torch.bfloat16)
bT2 = bS1 * bS2
print(f"torch-bf16 S1@(678,584) = {bS1[R, C].item():.6f}")
print(f"torch-bf16 T2@(678,584) = {bT2[R, C].item():.6f}")
