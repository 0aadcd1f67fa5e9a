"""Context measurement: torch.matmul (hipBLASLt) vs the hand-written MFMA
GEMM on identical bf16 random operands. Honest framing for profiles/:
the library is the upper envelope of tuned asm kernels; the gap shows
what inline-asm scheduling is worth on this chip."""
import os, sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext
ext = require_ext()

def t(fn, it, w=3):
    for _ in range(w):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it

for n, it in ((8192, 10), (16384, 4), (32768, 2)):
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    b = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = b.t().contiguous()
    c32 = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    o = torch.empty(n, n, dtype=torch.bfloat16, device="cuda")
    dt = t(lambda: torch.matmul(a, b, out=o), it)
    lib = 2 * n ** 3 / dt / 1e12
    dt2 = t(lambda: ext.gemm_bt_acc(c32, a, bt), it)
    ours = 2 * n ** 3 / dt2 / 1e12
    print(f"n={n}: hipBLASLt(torch.matmul)={lib:.0f} TF  ours(MFMA HIP)={ours:.0f} TF  ratio={ours/lib:.2f}")
    del a, b, bt, c32, o
