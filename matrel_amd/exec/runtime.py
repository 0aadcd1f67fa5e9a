"""Distributed runtime: one process per GPU over RCCL (torch.distributed).

The reference rides on Spark's driver/executor substrate;
here the substrate is torch.distributed with the "nccl" backend (= RCCL on
ROCm) over xGMI, one rank per MI355X, plus "gloo" for CPU-only test runs.

The Runtime owns:
  * the global process group and the row/column sub-groups of the logical
    (Pr x Pc) device grid (SUMMA broadcasts ride on these),
  * the compute stream of this rank (comm/compute overlap rides on
    ProcessGroupNCCL's internal RCCL stream: async_op collectives enqueue
    there and wait() fences the compute stream - see exec/summa.py),
  * device placement for this rank.

Single-process use (no init_process_group) is fully supported: world size 1,
grid (1,1) - every collective degenerates to a no-op. This is how the CPU
config-1 path (BASELINE.json configs[0]) and most unit tests run.
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from ..config import get_config


def _most_square_grid(p: int) -> Tuple[int, int]:
    """Factor p into (Pr, Pc) with Pr >= Pc, as square as possible.

    8 -> (4, 2): on one MI355X node the xGMI fabric is all-to-all
    point-to-point (7 links/GPU), so any factoring has full-bandwidth
    row/col groups; squarer grids minimize SUMMA panel traffic
    (panel bytes ~ N^2/Pr + N^2/Pc).
    """
    best = (p, 1)
    r = int(p ** 0.5)
    for pr in range(r, 0, -1):
        if p % pr == 0:
            a, b = max(pr, p // pr), min(pr, p // pr)
            best = (a, b)
            break
    return best


class Runtime:
    def __init__(self) -> None:
        self.rank: int = 0
        self.world_size: int = 1
        self.grid: Tuple[int, int] = (1, 1)
        self.device: torch.device = torch.device("cpu")
        self.backend: str = ""
        self._row_groups: List[Optional[dist.ProcessGroup]] = []
        self._col_groups: List[Optional[dist.ProcessGroup]] = []
        self.compute_stream: Optional[torch.cuda.Stream] = None
        self._initialized = False

    # --- init -----------------------------------------------------------

    def init(self, backend: Optional[str] = None,
             grid: Optional[Tuple[int, int]] = None) -> "Runtime":
        """Initialize from torchrun env vars if present, else single-process.

        backend: "nccl" (RCCL over xGMI; requires GPUs), "gloo" (CPU test
        path), or None to auto-select.
        """
        cfg = get_config()
        if dist.is_initialized():
            # torch.distributed was already set up by the launcher/test.
            self.rank = dist.get_rank()
            self.world_size = dist.get_world_size()
            self.backend = dist.get_backend()
        elif "RANK" in os.environ and "WORLD_SIZE" in os.environ and \
                int(os.environ["WORLD_SIZE"]) > 1:
            if backend is None:
                backend = "nccl" if torch.cuda.is_available() else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            dist.init_process_group(backend=backend)
            self.rank = dist.get_rank()
            self.world_size = dist.get_world_size()
            self.backend = backend
        else:
            self.rank, self.world_size, self.backend = 0, 1, ""

        if torch.cuda.is_available():
            local_rank = int(os.environ.get("LOCAL_RANK", self.rank))
            torch.cuda.set_device(local_rank % torch.cuda.device_count())
            self.device = torch.device("cuda", torch.cuda.current_device())
            self.compute_stream = torch.cuda.current_stream()
        else:
            self.device = torch.device("cpu")

        self.grid = grid or cfg.grid or _most_square_grid(self.world_size)
        pr, pc = self.grid
        if pr * pc != self.world_size:
            raise ValueError(
                f"grid {self.grid} does not match world size {self.world_size}")

        # Row/col sub-groups of the device grid for SUMMA broadcasts.
        # Rank layout is row-major: rank = i * Pc + j.
        self._row_groups, self._col_groups = [], []
        if self.world_size > 1:
            for i in range(pr):
                ranks = [i * pc + j for j in range(pc)]
                self._row_groups.append(dist.new_group(ranks))
            for j in range(pc):
                ranks = [i * pc + j for i in range(pr)]
                self._col_groups.append(dist.new_group(ranks))
        self._initialized = True
        return self

    # --- grid coordinates ----------------------------------------------

    @property
    def grid_row(self) -> int:
        return self.rank // self.grid[1]

    @property
    def grid_col(self) -> int:
        return self.rank % self.grid[1]

    def rank_of(self, i: int, j: int) -> int:
        return i * self.grid[1] + j

    def row_group(self, i: int):
        return self._row_groups[i] if self._row_groups else None

    def col_group(self, j: int):
        return self._col_groups[j] if self._col_groups else None

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    def barrier(self) -> None:
        if self.is_distributed:
            dist.barrier()

    def shutdown(self) -> None:
        if dist.is_initialized():
            dist.destroy_process_group()
        self._initialized = False


_runtime: Optional[Runtime] = None


def get_runtime() -> Runtime:
    global _runtime
    if _runtime is None:
        _runtime = Runtime().init()
    return _runtime


def init_runtime(backend: Optional[str] = None,
                 grid: Optional[Tuple[int, int]] = None) -> Runtime:
    global _runtime
    _runtime = Runtime().init(backend=backend, grid=grid)
    return _runtime


def reset_runtime() -> None:
    global _runtime
    if _runtime is not None:
        _runtime.shutdown()
    _runtime = None
