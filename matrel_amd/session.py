"""MatrelSession: the engine entry point.

Reference equivalent: MatfastSession (wraps SparkSession, registers the
matrix optimizer + strategies - SURVEY.md §2.1 "Session / API entry",
§3.1). Here the session owns the distributed runtime (one process per GPU,
RCCL over xGMI) and is the factory for Matrix handles.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from .blocks.matrix import BlockMatrix
from .blocks.schemes import Scheme
from .config import MatrelConfig, get_config, set_config
from .dsl import Matrix
from .exec.runtime import init_runtime
from .io.format import load_coo_text, load_matrix


class MatrelSession:
    def __init__(self, config: Optional[MatrelConfig] = None,
                 backend: Optional[str] = None,
                 grid: Optional[Tuple[int, int]] = None) -> None:
        if config is not None:
            set_config(config)
        self.config = get_config()
        self.runtime = init_runtime(backend=backend, grid=grid)

    # --- factories ------------------------------------------------------

    def randn(self, m: int, n: int, seed: int = 0, **kw) -> Matrix:
        return Matrix.from_blocks(BlockMatrix.randn(m, n, seed=seed, **kw))

    def rand(self, m: int, n: int, seed: int = 0, **kw) -> Matrix:
        return Matrix.from_blocks(BlockMatrix.rand(m, n, seed=seed, **kw))

    def zeros(self, m: int, n: int, **kw) -> Matrix:
        return Matrix.from_blocks(BlockMatrix.zeros(m, n, **kw))

    def ones(self, m: int, n: int, **kw) -> Matrix:
        import torch
        bm = BlockMatrix._filled(m, n, lambda s, g: torch.ones(*s), None,
                                 **kw)
        return Matrix.from_blocks(bm)

    def eye(self, n: int, **kw) -> Matrix:
        """Identity: only diagonal blocks materialized (block-sparse)."""
        import torch
        bm = BlockMatrix(n, n, **kw)
        for (bi, bj) in bm.owned_indices():
            if bi != bj:
                continue
            h, w = bm.block_shape(bi, bj)
            t = torch.eye(h, w)
            bm.blocks[(bi, bj)] = t.to(device=bm.device, dtype=bm.dtype)
        return Matrix.from_blocks(bm)

    def from_dense(self, t: torch.Tensor, **kw) -> Matrix:
        return Matrix.from_blocks(BlockMatrix.from_dense(t, **kw))

    def block_sparse_random(self, m: int, n: int, block_density: float,
                            seed: int = 0, **kw) -> Matrix:
        bm = BlockMatrix.block_sparse_random(m, n, block_density,
                                             seed=seed, **kw)
        return Matrix.from_blocks(bm, density=block_density)

    def sparse_random(self, m: int, n: int, density: float,
                      seed: int = 0, **kw) -> Matrix:
        """Element-level sparse random matrix (per-block CSR storage)."""
        bm = BlockMatrix.sparse_random(m, n, density, seed=seed, **kw)
        return Matrix.from_blocks(bm, density=density)

    def load(self, path: str, scheme: Optional[Scheme] = None) -> Matrix:
        bm = load_matrix(path, scheme=scheme)
        return Matrix.from_blocks(
            bm, density=getattr(bm, "_density_hint", None))

    def load_coo(self, path: str, nrows: int, ncols: int, **kw) -> Matrix:
        return Matrix.from_blocks(load_coo_text(path, nrows, ncols, **kw))

    # --- info -----------------------------------------------------------

    @property
    def world_size(self) -> int:
        return self.runtime.world_size

    @property
    def rank(self) -> int:
        return self.runtime.rank

    def metrics_report(self) -> dict:
        from .exec import metrics
        return metrics.report()

    def reset_metrics(self) -> None:
        from .exec import metrics
        metrics.reset()

    def shutdown(self) -> None:
        from .exec import tracing
        if self.config.trace:
            tracing.dump_trace()
        self.runtime.shutdown()
