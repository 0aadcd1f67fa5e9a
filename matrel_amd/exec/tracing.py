"""Operator-level tracing -> chrome-trace JSON.

Reference equivalent: the Spark web UI / event log stage timings
(SURVEY.md §5 "Tracing / profiling"). Two tracks per process:

  * tid 0 "host": per-op host wall timers (the executor's dispatch span).
  * tid 1 "gpu": per-op GPU spans from hipEvent pairs (torch.cuda.Event)
    recorded on the current stream around each op's kernel launches -
    resolved against a base event at dump time, so GPU time is
    attributed per op WITHOUT MATREL_TRACE_SYNC's serializing
    synchronize() calls. Host and GPU spans share the timeline, so
    overlap (async collectives / kernels finishing after the host moved
    on) is visible directly in chrome://tracing / Perfetto.

Kernel-level ground truth still comes from rocprofv3 (profiles/).
"""

from __future__ import annotations

import json
import os
import time
from contextlib import contextmanager
from typing import Any, Dict, List

import torch

from ..config import get_config

_events: List[Dict[str, Any]] = []
_gpu_pairs: List[tuple] = []  # (name, start_ev, end_ev, args)
_base: tuple = ()             # (base_event, host_ts_us)
_enabled_override = None


def enable(flag: bool = True) -> None:
    global _enabled_override
    _enabled_override = flag


def _enabled() -> bool:
    if _enabled_override is not None:
        return _enabled_override
    return get_config().trace


def _gpu_tracing() -> bool:
    return torch.cuda.is_available() and \
        os.environ.get("MATREL_TRACE_GPU", "1") != "0"


def _ensure_base() -> None:
    global _base
    if not _base:
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        ev.synchronize()
        _base = (ev, time.perf_counter_ns() / 1000.0)


@contextmanager
def trace_op(name: str, **args):
    if not _enabled():
        yield
        return
    sync = os.environ.get("MATREL_TRACE_SYNC") == "1"
    gpu = _gpu_tracing()
    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    ev0 = ev1 = None
    if gpu:
        _ensure_base()
        ev0 = torch.cuda.Event(enable_timing=True)
        ev0.record()
    t0 = time.perf_counter_ns()
    try:
        yield
    finally:
        if gpu:
            ev1 = torch.cuda.Event(enable_timing=True)
            ev1.record()
            _gpu_pairs.append((name, ev0, ev1,
                               {k: str(v) for k, v in args.items()}))
        if sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        t1 = time.perf_counter_ns()
        _events.append({
            "name": name, "ph": "X", "pid": os.getpid(), "tid": 0,
            "ts": t0 / 1000.0, "dur": (t1 - t0) / 1000.0,
            "args": {k: str(v) for k, v in args.items()},
        })


def _resolve_gpu_events() -> None:
    """Turn recorded hipEvent pairs into trace spans (called at dump)."""
    if not _gpu_pairs:
        return
    torch.cuda.synchronize()
    base_ev, base_us = _base
    for name, ev0, ev1, args in _gpu_pairs:
        # elapsed_time is ms between events on the stream timeline
        start_us = base_us + base_ev.elapsed_time(ev0) * 1000.0
        dur_us = max(0.0, ev0.elapsed_time(ev1) * 1000.0)
        _events.append({
            "name": f"{name} [gpu]", "ph": "X", "pid": os.getpid(),
            "tid": 1, "ts": start_us, "dur": dur_us, "args": args,
        })
    _gpu_pairs.clear()


def dump_trace(path: str = "") -> str:
    path = path or get_config().trace_path
    _resolve_gpu_events()
    meta = [
        {"ph": "M", "pid": os.getpid(), "tid": 0, "name": "thread_name",
         "args": {"name": "host"}},
        {"ph": "M", "pid": os.getpid(), "tid": 1, "name": "thread_name",
         "args": {"name": "gpu (hipEvent)"}},
    ]
    with open(path, "w") as f:
        json.dump({"traceEvents": meta + _events}, f)
    return path


def clear() -> None:
    global _base
    _events.clear()
    _gpu_pairs.clear()
    _base = ()
