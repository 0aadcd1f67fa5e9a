"""Failure handling: fail-fast on collective errors + checkpointed loops.

Reference equivalent: Spark gives MatRel lineage-based recompute and task
retry for free (SURVEY.md §5 "Failure detection"). On a fixed single-node
8-GPU RCCL job the right design is the opposite - fail fast and restart
from a checkpoint (no elasticity):

  * fail_fast(): wraps a callable; any RuntimeError from RCCL / HIP tears
    the whole job down immediately (os._exit) so torchrun's supervisor
    sees a clean non-zero exit instead of a hung communicator.
  * checkpointed_loop(): iterate a step function, saving named matrices
    every `every` steps via the on-disk block format; on restart, resume
    from the last complete checkpoint (save is atomic via tmp+rename of
    the manifest directory marker).
"""

from __future__ import annotations

import json
import os
import sys
import traceback
from typing import Callable, Dict

from ..blocks.matrix import BlockMatrix
from ..io.format import load_matrix, save_matrix
from .runtime import get_runtime


def fail_fast(fn: Callable, *args, **kw):
    """Run fn; on any error print, tear down the process group and exit
    non-zero immediately (a hung RCCL communicator is worse than a crash).
    """
    try:
        return fn(*args, **kw)
    except BaseException:
        traceback.print_exc()
        sys.stderr.write("matrel_amd: fail-fast teardown\n")
        sys.stderr.flush()
        os._exit(13)


class CheckpointedLoop:
    """Restartable iteration: state is a dict name -> BlockMatrix."""

    def __init__(self, path: str, every: int = 10) -> None:
        self.path = path
        self.every = every
        os.makedirs(path, exist_ok=True)

    def _meta_path(self) -> str:
        return os.path.join(self.path, "LOOP_STATE.json")

    def resume(self) -> (int, Dict[str, BlockMatrix]):
        """Return (start_step, state) - (0, {}) when no checkpoint."""
        mp = self._meta_path()
        if not os.path.exists(mp):
            return 0, {}
        with open(mp) as f:
            meta = json.load(f)
        state = {}
        for name in meta["names"]:
            state[name] = load_matrix(os.path.join(self.path, name))
        return meta["step"], state

    def save(self, step: int, state: Dict[str, BlockMatrix]) -> None:
        rt = get_runtime()
        for name, m in state.items():
            save_matrix(m, os.path.join(self.path, name))
        if rt.rank == 0:
            tmp = self._meta_path() + ".tmp"
            with open(tmp, "w") as f:
                json.dump({"step": step, "names": list(state)}, f)
            os.replace(tmp, self._meta_path())  # atomic commit
        rt.barrier()

    def run(self, step_fn: Callable[[int, Dict[str, BlockMatrix]],
                                    Dict[str, BlockMatrix]],
            n_steps: int) -> Dict[str, BlockMatrix]:
        start, state = self.resume()
        for step in range(start, n_steps):
            state = fail_fast(step_fn, step, state)
            if (step + 1) % self.every == 0 or step + 1 == n_steps:
                self.save(step + 1, state)
        return state
