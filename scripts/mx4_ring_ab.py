"""Race screen + A/B for the mx4 triple-buffered ring kernel (v3) vs
the double-buffered default (v1). Same quantized inputs => v3 must
match v1 EXACTLY (identical arithmetic and accumulation order)."""
import ctypes
import os, sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops import blockops
from matrel_amd.ops.backend import require_ext
ext = require_ext()
libc = ctypes.CDLL(None)

def setv(v):
    # the launcher caches the env ONCE (static); flip via re-exec model:
    # instead call the raw setter if present, else subprocess per variant
    os.environ["MATREL_MX4_V"] = str(v)

def run_case(n, v, seed):
    torch.manual_seed(seed)
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a4, asc = blockops.to_mx4(a)
    bt4, btsc = blockops.to_mx4(bt)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    ext.gemm_bt_mx4_acc(c, a4, asc, bt4, btsc, True)
    torch.cuda.synchronize()
    return a4, asc, bt4, btsc, c

if __name__ == "__main__":
    v = int(os.environ.get("MATREL_MX4_V", "1"))
    n = int(os.environ.get("RING_N", "4096"))
    reps = int(os.environ.get("RING_REPS", "3"))
    outs = []
    torch.manual_seed(11)
    a = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, n, device="cuda").to(torch.bfloat16)
    a4, asc = blockops.to_mx4(a)
    bt4, btsc = blockops.to_mx4(bt)
    c = torch.zeros(n, n, dtype=torch.float32, device="cuda")
    for r in range(reps):
        ext.gemm_bt_mx4_acc(c, a4, asc, bt4, btsc, True)
        torch.cuda.synchronize()
        outs.append(c.clone())
    for r in range(1, reps):
        assert torch.equal(outs[0], outs[r]), f"v{v} n={n} nondeterministic run {r}"
    torch.save(outs[0].cpu(), f"/tmp/ring_v{v}_{n}.pt")
    # perf
    if n >= 8192:
        ext.gemm_bt_mx4_acc(c, a4, asc, bt4, btsc, True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        it = 4
        for _ in range(it):
            ext.gemm_bt_mx4_acc(c, a4, asc, bt4, btsc, True)
        torch.cuda.synchronize()
        tf = 2 * n ** 3 / ((time.perf_counter() - t0) / it) / 1e12
        print(f"v{v} n={n}: {tf:.0f} TF", flush=True)
    print(f"v{v} n={n}: screen ok", flush=True)
