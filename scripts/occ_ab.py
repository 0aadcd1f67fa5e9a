import os, sys, time
import torch
sys.path.insert(0, "/root/repo")
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
for pad in (0, 40, 96):
    os.environ["MATREL_MX_LDS_PAD"] = str(pad)
    xs = [bench() for _ in range(3)]
    lds = 64 + pad
    wgs = 160 // lds
    print(f"mx4 LDS={lds}K ({wgs} WG/CU): " +
          " ".join(f"{x:.0f}" for x in xs), flush=True)
