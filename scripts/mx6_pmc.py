"""MX-fp6 GEMM dispatch loop for rocprofv3 PMC / kernel-trace runs.
Usage: GEMM_N=16384 python scripts/mx6_pmc.py"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops import blockops
from matrel_amd.ops.backend import require_ext

ext = require_ext()
n = int(os.environ.get("GEMM_N", "16384"))
a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
a6, asc = blockops.to_mx6(a)
bt6, btsc = blockops.to_mx6(bt)
c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
for _ in range(5):
    ext.gemm_bt_mx6_acc(c, a6, asc, bt6, btsc, True)
torch.cuda.synchronize()
print("done mx6", n)
