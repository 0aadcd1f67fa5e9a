#!/usr/bin/env python3
"""Flagship benchmark: N x N distributed dense bf16 matmul (BASELINE.json
metric: "TFLOPS + wall-clock, NxN distributed dense matmul at 1/2/4/8
MI355X").

Per-GPU problem size follows the BASELINE configs: 65536 at 1 GPU
(config 2) up to 262144 at 8 GPUs (config 3, SUMMA over xGMI); in between
N grows ~ g^(2/3) rounded to whole 16384-blocks. The matmul runs through
the full engine path (DSL -> optimizer -> physical plan -> SUMMA executor
-> hand-written MFMA GEMM), on synthetic random bf16 matrices.

Launch (the driver does this):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...

Rank 0 prints ONE JSON line with the whole-job aggregate TFLOPS.
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist

# N per GPU count: 1 -> config 2 (65k), 8 -> config 3 (256k)
N_TABLE = {1: 65536, 2: 98304, 4: 163840, 8: 262144}
CPU_N_TABLE = {1: 1024, 2: 2048, 4: 2048, 8: 4096}  # no-GPU smoke path


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--n-size", type=int, default=0,
                    help="override matrix side N")
    ap.add_argument("--block-size", type=int, default=0,
                    help="0 = auto (single block at 1 GPU, 16384 for SUMMA)")
    ap.add_argument("--precision", choices=["bf16", "mx8", "mx6", "mx4"],
                    default="bf16",
                    help="GEMM compute precision (mx8 = MX-scaled fp8; "
                         "NOT the headline config - BASELINE is bf16)")
    args = ap.parse_args()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    has_gpu = torch.cuda.is_available()

    import matrel_amd as ma

    block = args.block_size if has_gpu else 512
    ma.set_config(ma.MatrelConfig(block_size=block,
                                  gemm_precision=args.precision))
    sess = ma.MatrelSession()
    rt = sess.runtime
    world = rt.world_size
    assert world == args.gpus or args.gpus == 1, \
        f"world {world} != --gpus {args.gpus}"

    n = args.n_size or (N_TABLE if has_gpu else CPU_N_TABLE).get(
        world, 65536)

    A = sess.randn(n, n, seed=1)
    B = sess.randn(n, n, seed=2)

    resolved_block = A.plan.data.block_size if hasattr(A.plan, "data") \
        else block

    def step():
        C = (A @ B).compute()
        return C

    def sync():
        if has_gpu:
            torch.cuda.synchronize()
        if rt.is_distributed:
            dist.barrier()
        if has_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (RCCL needs device tensors; gloo needs CPU)
    if rt.is_distributed:
        dev = rt.device if rt.backend == "nccl" else torch.device("cpu")
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    flops = 2.0 * (float(n) ** 3)
    tflops = flops / (elapsed / args.steps) / 1e12

    if rt.rank == 0:
        out = {
            "metric": "dense_matmul_tflops",
            "value": round(tflops, 2),
            "unit": "TFLOPS",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": ({"mx8": "fp8_mx", "mx6": "fp6_mx", "mx4": "fp4_mx"}.get(args.precision, "bf16"))
                     if has_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": "NxN distributed dense matmul (BASELINE configs 2/3)",
                "n": n,
                "block_size": resolved_block,
                "global_batch": 1,
                "seq_len": n,
                "parallelism": f"summa_grid{rt.grid[0]}x{rt.grid[1]}",
            },
        }
        print(json.dumps(out))
    rt.shutdown()


if __name__ == "__main__":
    main()
