"""Single-variant GEMM dispatch loop for rocprofv3 PMC runs.
Usage: GEMM_VARIANT=1|2 GEMM_N=8192 python scripts/gemm_pmc.py"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext  # noqa

ext = require_ext()
v = int(os.environ.get("GEMM_VARIANT", "1"))
n = int(os.environ.get("GEMM_N", "8192"))
ext.set_gemm_variant(v)
a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
for _ in range(5):
    ext.gemm_bt_acc(c, a, bt)
torch.cuda.synchronize()
print("done", v, n)
