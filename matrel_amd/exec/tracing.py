"""Operator-level tracing -> chrome-trace JSON.

Reference equivalent: the Spark web UI / event log stage timings
(SURVEY.md §5 "Tracing / profiling"). Here: per-op host timers (plus
torch.cuda.synchronize bracketing when MATREL_TRACE_SYNC=1 so GPU time is
attributed to the op that launched it), dumped as a chrome://tracing /
Perfetto-compatible JSON via `dump_trace()` or automatically when
MatrelConfig.trace is set. Kernel-level ground truth comes from rocprofv3
(profiles/), not from this module.
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
_enabled_override = None


def enable(flag: bool = True) -> None:
    global _enabled_override
    _enabled_override = flag


def _enabled() -> bool:
    if _enabled_override is not None:
        return _enabled_override
    return get_config().trace


@contextmanager
def trace_op(name: str, **args):
    if not _enabled():
        yield
        return
    sync = os.environ.get("MATREL_TRACE_SYNC") == "1"
    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter_ns()
    try:
        yield
    finally:
        if sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        t1 = time.perf_counter_ns()
        _events.append({
            "name": name, "ph": "X", "pid": os.getpid(), "tid": 0,
            "ts": t0 / 1000.0, "dur": (t1 - t0) / 1000.0,
            "args": {k: str(v) for k, v in args.items()},
        })


def dump_trace(path: str = "") -> str:
    path = path or get_config().trace_path
    with open(path, "w") as f:
        json.dump({"traceEvents": _events}, f)
    return path


def clear() -> None:
    _events.clear()
