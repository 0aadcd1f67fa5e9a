"""Mixed-precision subspace iteration with the MX GEMM tiers.

Estimates the dominant k-dim subspace of a square matrix by block
power iteration, running the BULK of the work at a quantized MX tier
(gemm_precision="mx6": fp8-class accuracy at ~2x the bf16 GEMM rate
and 3/8 the operand bytes), then polishing with ONE bf16 iteration.

This is the intended use of the mx4/mx6 tiers (docs/NUMERICS.md):
the quantized conversion of A is computed ONCE and cached on the
BlockMatrix (exec/summa.py _mx8_cached), so every subsequent multiply
by A reads 3/8 the HBM/xGMI bytes and runs on the double-rate MFMA
path - iterative methods that REUSE an operand are exactly where the
tiers pay. One-shot products (e.g. a single OLS Gram pass) are
conversion-dominated and should stay bf16.

Run:  python examples/mixed_precision_refinement.py [--n N --iters K]
(On CPU the MX tiers are unavailable; the example then runs both
phases at the default precision to demonstrate the control flow.)
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa

import matrel_amd as ma  # noqa


def orthonormalize(xt: torch.Tensor) -> torch.Tensor:
    q, _ = torch.linalg.qr(xt.float())
    return q


def iterate(s, A, x_t, iters):
    """iters block power steps A @ X under the CURRENT gemm precision.
    Between steps only the (cheap) per-column norms are rescaled; the
    full QR re-orthonormalization runs once at phase end - the standard
    subspace-iteration economy, and it keeps each step GEMM-dominated."""
    dbg = os.environ.get("MPR_DEBUG") == "1"
    for _ in range(iters):
        it0 = time.perf_counter()
        X = s.from_dense(x_t.to(torch.bfloat16))
        Y = (A @ X).compute()
        y = Y.to_tensor().float()
        x_t = y / (y.norm(dim=0, keepdim=True) + 1e-20)
        if dbg:
            torch.cuda.synchronize()
            print(f"    it {time.perf_counter()-it0:.4f}s", flush=True)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return x_t  # caller re-orthonormalizes (QR costs more than a step)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=0)
    ap.add_argument("--iters", type=int, default=6)
    args = ap.parse_args()
    gpu = torch.cuda.is_available()
    n = args.n or (32768 if gpu else 1024)
    dim = 1024 if gpu else 128
    bs = 16384 if gpu else 256
    mx = "mx6" if gpu else "bf16"

    ma.set_config(ma.MatrelConfig(block_size=bs, gemm_precision=mx))
    s = ma.MatrelSession()
    A = s.randn(n, n, seed=1)
    g = torch.Generator().manual_seed(2)
    x0 = orthonormalize(torch.randn(n, dim, generator=g))
    if gpu:
        x0 = x0.cuda()

    # warm both paths once (JIT, caches, the one-time mx conversion)
    _ = iterate(s, A, x0, 1)

    t0 = time.perf_counter()
    x_mx = iterate(s, A, x0, args.iters)
    t_mx = time.perf_counter() - t0
    x_mx = orthonormalize(x_mx)

    # polish: ONE bf16 iteration removes the quantization bias
    ma.get_config().gemm_precision = "bf16"
    _ = iterate(s, A, x_mx, 1)  # warm bf16 path
    t0 = time.perf_counter()
    x_fin = iterate(s, A, x_mx, 1)
    t_pol = time.perf_counter() - t0
    x_fin = orthonormalize(x_fin)

    # reference: the same total iteration count, all bf16
    _ = iterate(s, A, x0, 1)  # warm (first call after a phase boundary
    # pays one-time library/allocator setup, ~0.4 s - not a step cost)
    t0 = time.perf_counter()
    x_ref = iterate(s, A, x0, args.iters)
    t_ref = time.perf_counter() - t0
    x_ref = orthonormalize(iterate(s, A, orthonormalize(x_ref), 1))

    def residual(x_t):
        """Rayleigh residual ||AX - X(X^T AX)||_F / ||AX||_F: the
        invariant-subspace quality of X, independent of which of the
        near-degenerate trailing directions a particular run picked."""
        X = s.from_dense(x_t.to(torch.bfloat16))
        y = (A @ X).compute().to_tensor().float()
        xf = x_t.float()
        r = y - xf @ (xf.T @ y)
        return (r.norm() / y.norm()).item()

    if s.rank == 0:
        # subspace distance: principal angles via singular values of
        # Q_mx^T Q_ref (1 - smallest sv == largest angle's cos gap)
        res_mx = residual(x_fin)
        res_bf = residual(x_ref)
        sv = torch.linalg.svdvals(x_fin.float().T @ x_ref.float())
        mean_gap = (1 - sv).mean().item()
        per_it_mx = t_mx / args.iters * 1e3
        per_it_bf = t_ref / args.iters * 1e3
        print(f"[mixed-precision subspace iteration] n={n} dim={dim} "
              f"iters={args.iters} tier={mx}")
        print(f"  {mx} per-iteration:  {per_it_mx:7.1f} ms")
        print(f"  bf16 per-iteration: {per_it_bf:7.1f} ms  "
              f"(speedup {per_it_bf / per_it_mx:.2f}x)")
        print(f"  bf16 polish:        {t_pol*1e3:7.1f} ms (one step)")
        print(f"  invariant-subspace residual: mixed {res_mx:.4f} "
              f"vs all-bf16 {res_bf:.4f} (subspace mean gap "
              f"{mean_gap:.1e})")
        if mx != "bf16":
            assert res_mx < res_bf * 1.2 + 1e-3, (res_mx, res_bf)
            print("  quantized bulk + one bf16 polish converges as "
                  "well as all-bf16")
    s.shutdown()


if __name__ == "__main__":
    main()
