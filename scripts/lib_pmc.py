"""torch.matmul (hipBLASLt) dispatch loop for rocprofv3 PMC comparison
against scripts/gemm_pmc.py (same shape, same dispatch count)."""
import os
import sys

import torch

n = int(os.environ.get("GEMM_N", "16384"))
a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
b = torch.randn(n, n, device="cuda").to(torch.bfloat16)
c = torch.empty(n, n, device="cuda", dtype=torch.bfloat16)
for _ in range(5):
    torch.matmul(a, b, out=c)
torch.cuda.synchronize()
print("done lib", n)
