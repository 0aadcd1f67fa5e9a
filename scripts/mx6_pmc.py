"""MX-quantized GEMM dispatch loop for rocprofv3 PMC / kernel-trace runs.
Usage: MX_PREC=mx6 GEMM_N=16384 python scripts/mx6_pmc.py"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops import blockops
from matrel_amd.ops.backend import require_ext

ext = require_ext()
n = int(os.environ.get("GEMM_N", "16384"))
prec = os.environ.get("MX_PREC", "mx6")
conv = {"mx8": blockops.to_mx8, "mx6": blockops.to_mx6,
        "mx4": blockops.to_mx4}[prec]
gemm = {"mx8": ext.gemm_bt_mx8_acc, "mx6": ext.gemm_bt_mx6_acc,
        "mx4": ext.gemm_bt_mx4_acc}[prec]
a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
a6, asc = conv(a)
bt6, btsc = conv(bt)
c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
for _ in range(5):
    gemm(c, a6, asc, bt6, btsc, True)
torch.cuda.synchronize()
print("done", prec, n)
