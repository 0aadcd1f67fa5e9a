"""Perf-probe A/B: mx6 kernel with vs without the H (regs 4-5) b64 LDS
read stream. NOHI results are numerically WRONG - this isolates the
cost of the second read stream in the otherwise-identical kernel."""
import os, sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops import blockops
from matrel_amd.ops.backend import require_ext
ext = require_ext()
n = 16384
a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
a6, asc = blockops.to_mx6(a)
bt6, btsc = blockops.to_mx6(bt)
c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
def bench(it=4):
    ext.gemm_bt_mx6_acc(c, a6, asc, bt6, btsc, True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        ext.gemm_bt_mx6_acc(c, a6, asc, bt6, btsc, True)
    torch.cuda.synchronize()
    return 2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12
for mode in ("full", "nohi", "nostage"):
    if mode == "nohi":
        os.environ["MATREL_MX6_PROBE_NOHI"] = "1"
    elif mode == "nostage":
        os.environ["MATREL_MX6_PROBE_NOHI"] = "2"
    else:
        os.environ.pop("MATREL_MX6_PROBE_NOHI", None)
    xs = [bench() for _ in range(3)]
    print(f"mx6 {mode}: " + " ".join(f"{x:.0f}" for x in xs), flush=True)
