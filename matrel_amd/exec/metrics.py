"""Per-run metrics: op counts, wall-clock, estimated FLOPs and comm bytes.

Reference equivalent: the Spark metrics system / per-stage timings
(SURVEY.md §5 "Metrics / logging / observability"). The executor feeds this
collector; `report()` returns a JSON-serializable summary (also written to
MATREL_METRICS_PATH when set). This is host-side bookkeeping - kernel-level
ground truth comes from rocprofv3 (profiles/).
"""

from __future__ import annotations

import json
import os
import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Any, Dict

_ops: Dict[str, Dict[str, float]] = defaultdict(
    lambda: {"count": 0, "wall_s": 0.0, "est_flops": 0.0,
             "est_comm_bytes": 0.0})
_comm: Dict[str, Dict[str, float]] = defaultdict(
    lambda: {"count": 0, "bytes": 0.0})
_run_start = time.time()


def add_comm(kind: str, nbytes: int) -> None:
    """Record ACTUAL bytes handed to a collective (broadcast / all_reduce
    / send; called by summa.py and BlockMatrix.repartition)."""
    e = _comm[kind]
    e["count"] += 1
    e["bytes"] += float(nbytes)


@contextmanager
def record_op(name: str, est_flops: float = 0.0, est_comm_bytes: float = 0.0):
    t0 = time.perf_counter()
    try:
        yield
    finally:
        dt = time.perf_counter() - t0
        e = _ops[name]
        e["count"] += 1
        e["wall_s"] += dt
        e["est_flops"] += est_flops
        e["est_comm_bytes"] += est_comm_bytes


def report() -> Dict[str, Any]:
    total_flops = sum(e["est_flops"] for e in _ops.values())
    total_wall = sum(e["wall_s"] for e in _ops.values())
    rep = {
        "elapsed_s": time.time() - _run_start,
        "total_est_tflop": total_flops / 1e12,
        "op_wall_s": total_wall,
        "avg_tflops": (total_flops / total_wall / 1e12) if total_wall else 0,
        "ops": {k: dict(v) for k, v in sorted(_ops.items())},
        "collectives": {k: dict(v) for k, v in sorted(_comm.items())},
        "total_comm_gb": sum(v["bytes"] for v in _comm.values()) / 1e9,
    }
    path = os.environ.get("MATREL_METRICS_PATH")
    if path:
        with open(path, "w") as f:
            json.dump(rep, f, indent=1)
    return rep


def reset() -> None:
    global _run_start
    _ops.clear()
    _comm.clear()
    _run_start = time.time()
