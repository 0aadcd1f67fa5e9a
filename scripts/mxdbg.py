import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from matrel_amd.ops.backend import require_ext
ext = require_ext()

def case(M, N, K, seed, dist="int"):
    g = torch.Generator().manual_seed(seed)
    if dist == "int":
        a = torch.randint(-3, 4, (M, K), generator=g).to(torch.bfloat16).cuda()
        bt = torch.randint(-3, 4, (N, K), generator=g).to(torch.bfloat16).cuda()
    else:
        a = torch.randn(M, K, generator=g).to(torch.bfloat16).cuda()
        bt = torch.randn(N, K, generator=g).to(torch.bfloat16).cuda()
    a8, asc = ext.mx8_convert(a)
    bt8, btsc = ext.mx8_convert(bt)
    c = torch.empty(M, N, dtype=torch.float32, device="cuda")
    ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
    ref = a.float() @ bt.float().t()
    err = (c - ref).abs()
    thresh = 0.5 if dist == "int" else 0.35 * (K ** 0.5)
    bad = (err > thresh)
    print(f"M={M} N={N} K={K} {dist}: bad frac "
          f"{bad.float().mean().item():.4f} maxerr {err.max().item():.1f}"
          f" (thresh {thresh:.1f})")
    if bad.any():
        ys, xs = torch.nonzero(bad, as_tuple=True)
        i, j = ys[0].item(), xs[0].item()
        print("  first bad", i, j, "got", c[i, j].item(), "want",
              ref[i, j].item())
        b32 = bad.reshape(M // 32, 32, N // 32, 32).any(dim=(1, 3))
        print("  frag map:\n", b32.to(torch.int8).cpu().numpy())

case(256, 256, 64, 1)
case(512, 512, 512, 4)
case(256, 256, 64, 5, "randn")
case(256, 256, 512, 6, "randn")
case(512, 512, 512, 7, "randn")


def dequant(a8, sc):
    bits = a8.to(torch.int32)
    s = bits >> 7
    be = (bits >> 3) & 0xF
    m = bits & 7
    val = torch.where(be > 0, (2.0 ** (be - 7)) * (1 + m / 8.0),
                      (2.0 ** -6) * (m / 8.0))
    val = torch.where(s > 0, -val, val)
    if sc.dtype == torch.int32:
        # packed-dword layout [K/128][rows] (mxconvert round 2): byte b
        # of word t covers 32-block 4t+b
        kt, rows = sc.shape
        scb = sc.cpu().view(torch.uint8).reshape(kt, rows, 4) \
            .permute(0, 2, 1).reshape(kt * 4, rows).to(sc.device)
    else:
        scb = sc
    scale = 2.0 ** (scb.to(torch.int32).float().t() - 127)
    return val.float() * scale.repeat_interleave(32, dim=1)


g = torch.Generator().manual_seed(5)
M = N = 256
K = 64
a = torch.randn(M, K, generator=g).to(torch.bfloat16).cuda()
bt = torch.randn(N, K, generator=g).to(torch.bfloat16).cuda()
a8, asc = ext.mx8_convert(a)
bt8, btsc = ext.mx8_convert(bt)
c = torch.empty(M, N, dtype=torch.float32, device="cuda")
ext.gemm_bt_mx8_acc(c, a8, asc, bt8, btsc, True)
da = dequant(a8, asc)
db = dequant(bt8, btsc)
ref_q = da @ db.t()          # what the hardware SHOULD compute
ref_f = a.float() @ bt.float().t()
print("kernel vs dequant-product maxerr:",
      (c - ref_q).abs().max().item())
print("dequant-product vs fp32 maxerr:",
      (ref_q - ref_f).abs().max().item())
print("conv elementwise max rel:",
      ((da - a.float()).abs() / (a.float().abs() + 1e-3)).max().item())
bad = (c - ref_q).abs() > 0.5
if bad.any():
    b32 = bad.reshape(M // 32, 32, N // 32, 32).any(dim=(1, 3))
    print("kernel-vs-dequant frag map:\n", b32.to(torch.int8).cpu().numpy())


print("=== per-k-block localization (K=64: blocks 0,1) ===")
for b in range(2):
    a8m = a8.clone()
    bt8m = bt8.clone()
    for blk in range(2):
        if blk != b:
            a8m[:, blk * 32:(blk + 1) * 32] = 0
            bt8m[:, blk * 32:(blk + 1) * 32] = 0
    cb = torch.empty(M, N, dtype=torch.float32, device="cuda")
    ext.gemm_bt_mx8_acc(cb, a8m, asc, bt8m, btsc, True)
    dam = dequant(a8m, asc)
    dbm = dequant(bt8m, btsc)
    refb = dam @ dbm.t()
    print(f"block {b}: kernel-vs-dequant maxerr "
          f"{(cb - refb).abs().max().item():.4f}")

print("=== scale-uniform payload test (force all scales equal) ===")
asc_u = torch.full_like(asc, 121)
btsc_u = torch.full_like(btsc, 121)
cu = torch.empty(M, N, dtype=torch.float32, device="cuda")
ext.gemm_bt_mx8_acc(cu, a8, asc_u, bt8, btsc_u, True)
ref_u = dequant(a8, asc_u) @ dequant(bt8, btsc_u).t()
print("uniform scales: maxerr", (cu - ref_u).abs().max().item())

print("=== payload-uniform, true scales ===")
a8_u = torch.full_like(a8, 0x38)  # all 1.0
bt8_u = torch.full_like(bt8, 0x38)
cs = torch.empty(M, N, dtype=torch.float32, device="cuda")
ext.gemm_bt_mx8_acc(cs, a8_u, asc, bt8_u, btsc, True)
ref_s = dequant(a8_u, asc) @ dequant(bt8_u, btsc).t()
print("true scales, ones payload: maxerr",
      (cs - ref_s).abs().max().item(),
      " relerr", ((cs - ref_s).abs() / (ref_s.abs() + 1e-3)).max().item())
