"""hipGraph capture of a matrix expression (launch-bound plans).

MI355X-native replacement for a tracing compiler (repo design note: "HIP
streams and graphs instead of a tracing compiler"): a CompiledExpression
captures the full kernel sequence of one plan evaluation into a hipGraph
(torch.cuda.CUDAGraph == hipGraph on ROCm) and replays it with a single
launch. Pays off when the plan is many SMALL kernels (fine block grids,
sparse plans with thousands of tiny GEMMs - ~1.2-1.9 us saved per kernel
boundary); for 16k^2 blocks the kernels are ms-scale and eager launch is
already negligible.

Constraints (documented, enforced):
  * single process only (RCCL collectives are not captured here),
  * the expression's leaves must already be materialized on the GPU,
  * no host-side data-dependent control flow in the plan (Solve/Inverse
    gather to host -> not capturable).
Falls back to eager recomputation when capture is impossible.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..dsl import Matrix
from ..plan import logical as L
from ..plan.optimizer import optimize
from ..plan.physical import plan_physical
from .runtime import get_runtime

_UNCAPTURABLE = (L.Solve, L.Inverse, L.SemiJoinValue, L.Select, L.CrossJoin)


def _capturable(p: L.Plan) -> bool:
    if isinstance(p, _UNCAPTURABLE):
        return False
    return all(_capturable(c) for c in p.children)


class CompiledExpression:
    def __init__(self, m: Matrix) -> None:
        from .executor import Executor
        self.plan = optimize(m.plan)
        rt = get_runtime()
        self.eager = (rt.is_distributed or rt.device.type != "cuda"
                      or not _capturable(self.plan))
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.result: Optional[Matrix] = None
        if self.eager:
            return
        physical = plan_physical(self.plan)
        # warmup on a side stream (allocator state), then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            Executor(physical).execute(self.plan)
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        ex = Executor(plan_physical(self.plan))
        with torch.cuda.graph(self.graph):
            out = ex.execute(self.plan)
        self.result = Matrix(L.Leaf(out))

    def __call__(self) -> Matrix:
        """Replay: ONE hipGraph launch; the returned Matrix's blocks are
        the graph's output buffers (valid until the next replay)."""
        if self.eager:
            from .executor import Executor
            out = Executor().execute(self.plan)
            return Matrix(L.Leaf(out))
        self.graph.replay()
        return self.result


def compile_expression(m: Matrix) -> CompiledExpression:
    return CompiledExpression(m)
