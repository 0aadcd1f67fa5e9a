"""Minimal repro for the deterministic-mode fuzz mismatch
(seed 91, N=1088, iter 136)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import matrel_amd as ma  # noqa
from matrel_amd.exec.runtime import reset_runtime  # noqa

SPEC = ('max', ('shift', ('leaf', 3), -1.329),
        ('mul',
         ('sub', ('scale', ('leaf', 1), -1.412),
          ('sub', ('sel', ('leaf', 2), 453, 934),
           ('mul', ('leaf', 3), ('leaf', 3)))),
         ('sub', ('shift', ('mul', ('leaf', 1), ('leaf', 1)), -1.022),
          ('sub', ('leaf', 3), ('mm', ('leaf', 1), ('leaf', 3))))))

N = 1088


def build(spec, leaves):
    kind = spec[0]
    if kind == "leaf":
        m, t = leaves[spec[1]]
        return m, t.clone()
    if kind in ("add", "sub", "mul", "min", "max", "mm"):
        am, at = build(spec[1], leaves)
        bm, bt = build(spec[2], leaves)
        if kind == "add":
            return am + bm, at + bt
        if kind == "sub":
            return am - bm, at - bt
        if kind == "mul":
            return am * bm, at * bt
        if kind == "min":
            return am.elem_min(bm), torch.minimum(at, bt)
        if kind == "max":
            return am.elem_max(bm), torch.maximum(at, bt)
        return (am @ bm) * (1.0 / N), (at @ bt) / N
    if kind == "t":
        am, at = build(spec[1], leaves)
        return am.T, at.t()
    if kind == "scale":
        am, at = build(spec[1], leaves)
        return am * spec[2], at * spec[2]
    if kind == "shift":
        am, at = build(spec[1], leaves)
        return am + spec[2], at + spec[2]
    lo, hi = spec[2], spec[3]
    am, at = build(spec[1], leaves)
    ref = torch.zeros_like(at)
    ref[lo:hi] = at[lo:hi]
    return am.select_rows(lo, hi), ref


def run(det):
    reset_runtime()
    ma.set_config(ma.MatrelConfig(block_size=256, deterministic=det))
    s = ma.MatrelSession()
    leaves = []
    for seed in (1, 2, 3):
        m = s.randn(N, N, seed=seed)
        leaves.append((m, m.to_tensor().float()))
    sp = s.block_sparse_random(N, N, 0.5, seed=4)
    leaves.append((sp, sp.to_tensor().float()))
    m, ref = build(SPEC, leaves)
    out = m.compute().to_tensor().float()
    err = (out - ref).abs()
    tol = 0.15 + 0.05 * ref.abs()
    viol = err - tol
    idx = viol.argmax().item()
    r, c = idx // err.shape[1], idx % err.shape[1]
    nviol = int((viol > 0).sum().item())
    print(f"det={det}: maxerr {err.max().item():.4f}; violations "
          f"{nviol}; worst at ({r},{c}) out={out[r,c].item():.6f} "
          f"ref={ref[r,c].item():.6f} err={err[r,c].item():.4f}",
          flush=True)
    if nviol:
        vr, vc = (viol > 0).nonzero()[:8].tolist(), None
        print("  first violators:", vr, flush=True)
    reset_runtime()
    return int((viol > 0).sum().item())


if __name__ == "__main__":
    v1 = run(det=True)
    v0 = run(det=False)
    print(f"violations det={v1} nondet={v0}")
