"""Observability tour: plan inspection, chrome-trace, metrics report.

The reference inherits the Spark web UI (stages, tasks, shuffle bytes);
this engine's equivalents are `explain()`, the chrome-trace emitter and
the per-run JSON metrics (SURVEY.md §5 "Tracing / profiling").

Run: python examples/observability_tour.py
Then open /tmp/matrel_tour_trace.json in chrome://tracing or Perfetto.
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import matrel_amd as ma  # noqa

cfg = ma.MatrelConfig(block_size=256)
cfg.trace = True
cfg.trace_path = "/tmp/matrel_tour_trace.json"
ma.set_config(cfg)

s = ma.MatrelSession()
X = s.randn(2048, 512, seed=1)
y = s.randn(2048, 1, seed=2)

# 1. plan inspection: the OLS normal equations - watch the optimizer share
#    X^T (cse-shared), pick per-multiply algorithms and schemes
expr = (X.T @ X).inverse() @ (X.T @ y)
print(expr.explain(verbose=False))
print()

# 2. execute with tracing on; the chrome-trace lands at cfg.trace_path
w = expr.compute()
print("w[:4] =", w.to_tensor()[:4].reshape(-1).tolist())

# 3. per-run metrics: wall clock, estimated flops, bytes per collective
rep = s.metrics_report()
print(json.dumps(rep, indent=1, default=str)[:800])
s.shutdown()  # flushes the chrome-trace to cfg.trace_path
print(f"\nchrome trace: {cfg.trace_path}")
