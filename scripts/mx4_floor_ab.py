"""mx4 staging-vs-compute decomposition probe. HISTORICAL: the NOSTAGE
kernel path was removed after the measurement (full 2865 / floor 4358
TF at 16384^3, recorded in profiles/README.md \"ring-kernel study\");
without it this script now measures the full kernel twice."""
import os, sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops import blockops
from matrel_amd.ops.backend import require_ext
ext = require_ext()
n = 16384
a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
a4, asc = blockops.to_mx4(a)
bt4, btsc = blockops.to_mx4(bt)
c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
def bench(it=4):
    ext.gemm_bt_mx4_acc(c, a4, asc, bt4, btsc, True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        ext.gemm_bt_mx4_acc(c, a4, asc, bt4, btsc, True)
    torch.cuda.synchronize()
    return 2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12
for mode in ("full", "nostage"):
    if mode == "nostage":
        os.environ["MATREL_MX4_PROBE_NOSTAGE"] = "1"
    else:
        os.environ.pop("MATREL_MX4_PROBE_NOSTAGE", None)
    xs = [bench() for _ in range(3)]
    print(f"mx4 {mode}: " + " ".join(f"{x:.0f}" for x in xs), flush=True)
