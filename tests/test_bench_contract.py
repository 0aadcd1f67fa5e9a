"""Pin the driver's bench contract: bench.py main() at world 2 (gloo)
must produce ONE JSON line from rank 0 with the agreed fields - the
exact interface the round-end SCALE runs depend on."""

import io
import json
import os
import sys
from contextlib import redirect_stdout

import pytest
import torch.multiprocessing as mp


def _worker(rank: int, world: int, port: int, q) -> None:
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, root)
    import bench
    sys.argv = ["bench.py", "--gpus", str(world), "--steps", "1",
                "--warmup", "0", "--n-size", "512"]
    buf = io.StringIO()
    with redirect_stdout(buf):
        bench.main()
    q.put((rank, buf.getvalue()))


def test_bench_main_world2(tmp_path):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_worker, args=(2, 29952, q), nprocs=2, join=True)
    outs = dict(q.get() for _ in range(2))
    # exactly one rank prints; the line is valid JSON with the contract
    lines = [ln for r in outs.values() for ln in r.strip().splitlines()
             if ln.startswith("{")]
    assert len(lines) == 1, outs
    d = json.loads(lines[0])
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"):
        assert field in d, field
    assert d["n_gpus"] == 2
    assert d["metric"] == "dense_matmul_tflops"
    assert d["data"] == "synthetic"
    assert d["config"]["parallelism"].startswith("summa_grid")


def test_bench_main_world8(tmp_path):
    """The SCALE run's exact rank count (8, 4x2 grid) through bench
    main() on gloo - rendezvous, grid derivation, MAX-over-ranks
    reduction and the single rank-0 JSON line."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_worker, args=(8, 29954, q), nprocs=8, join=True)
    outs = dict(q.get() for _ in range(8))
    lines = [ln for r in outs.values() for ln in r.strip().splitlines()
             if ln.startswith("{")]
    assert len(lines) == 1, outs
    d = json.loads(lines[0])
    assert d["n_gpus"] == 8
    assert d["config"]["parallelism"] == "summa_grid4x2"


def test_bench_single_process(tmp_path):
    import subprocess
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = {k: v for k, v in os.environ.items()
           if k not in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
                        "MASTER_PORT")}
    out = subprocess.run(
        [sys.executable, os.path.join(root, "bench.py"), "--steps", "1",
         "--warmup", "0", "--n-size", "256"],
        capture_output=True, text=True, cwd=root, env=env, timeout=300)
    assert out.returncode == 0, out.stderr[-800:]
    lines = [ln for ln in out.stdout.strip().splitlines()
             if ln.startswith("{")]
    assert len(lines) == 1
    d = json.loads(lines[0])
    assert d["n_gpus"] == 1 and d["metric"] == "dense_matmul_tflops"
