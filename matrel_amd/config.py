"""Global configuration for the MatRel-AMD engine.

The reference (purduedb/MatRel, a Spark-SQL fork) configures block size and
partition counts via SparkConf keys (SURVEY.md §5 "Config / flag system").
Here a single dataclass covers the whole engine: block geometry, dtype,
device-grid shape, SUMMA panel size and debug switches. Values can be set
programmatically, via environment variables (MATREL_*), or per-session.

MI355X-first defaults: blocks are LARGE (multi-GB per GPU shard, not Spark's
1000x1000) because each GPU has 288 GB HBM3E and kernel-launch amortization
wants >= 8k-square tiles; the CPU default is small so the multi-block code
paths are exercised in CPU tests (BASELINE.json config 1: 1k x 1k, 2x2 grid).
"""

from __future__ import annotations

import dataclasses
import os
from typing import Optional, Tuple


def _env_int(name: str, default: int) -> int:
    v = os.environ.get(name)
    return int(v) if v else default


def _env_str(name: str, default: str) -> str:
    return os.environ.get(name, default)


@dataclasses.dataclass
class MatrelConfig:
    # --- block geometry ---
    # Side length of a square block. A distributed matrix of shape (m, n) is
    # tiled into ceil(m/bs) x ceil(n/bs) blocks. GPU default 16384 (bf16 block
    # = 512 MiB); CPU tests override to something small (e.g. 512).
    block_size: int = 16384

    # --- device grid ---
    # Shape (Pr, Pc) of the logical GPU grid used by grid partition schemes
    # and SUMMA. None -> derived from world size (most-square factoring,
    # e.g. 8 -> 4x2). Single process -> (1, 1).
    grid: Optional[Tuple[int, int]] = None

    # --- numerics ---
    # Compute dtype for dense block math on GPU ("bf16" MFMA path).
    # Accumulation is always fp32. CPU fallback computes in fp32.
    dtype: str = "bf16"
    # dtype used for reductions / solves that need extra precision.
    solve_dtype: str = "fp32"
    # Solve/Inverse gather their operand DENSE to every rank (the OLS k x k
    # Gram matrix). Refuse silently-catastrophic gathers above this side.
    solve_max_side: int = 32768
    # GEMM compute precision: "bf16" (default), "mx8" (MX-scaled fp8
    # e4m3, ~2x the bf16 MFMA rate at ~2-3 decimal digits), "mx6"
    # (MX fp6 e2m3: fp8-class accuracy at the fp4 rate) or "mx4"
    # (MX-scaled fp4 e2m1, ~2x the fp8 rate at ~1-2 significant digits -
    # screening-pass precision). All opt-in.
    gemm_precision: str = "bf16"

    # --- sparsity ---
    # A block whose density is below this threshold is stored sparse (CSR).
    sparse_threshold: float = 0.05
    # GPU compute routing: a CSR block at or above this density computes
    # faster densified on the MFMA pipe than through the CSR kernels
    # (measured crossover ~1% at 8-16k blocks - profiles/ spmm isolation
    # entry). Storage/wire stay CSR; only the local product densifies.
    spmm_densify_crossover: float = 0.01

    # --- execution ---
    # Use a separate HIP stream for collectives so comm overlaps compute.
    overlap_comm: bool = True
    # SUMMA prefetch depth: how many k-step panel broadcasts are kept in
    # flight AHEAD of the step being computed (RCCL runs them under the
    # compute). 1 = classic double buffering; >1 pays more panel memory
    # for deeper overlap on slow links; ignored when overlap_comm=False
    # (serialized debug mode). Panel SIZE is block_size; grid shape is
    # MATREL_GRID - together these are the driver-sweepable SUMMA knobs.
    summa_prefetch: int = 1
    # Owner-side SUMMA panel prep on a second HIP stream (overlaps the
    # previous step's GEMMs). Kill switch for triage: MATREL_PREP_STREAM=0.
    prep_stream: bool = True
    # Memory cap for replicating an operand in broadcast-MM. This is ONLY
    # a memory guard (replication must fit next to the shards in 288 GB
    # HBM3E); the bcast-vs-SUMMA choice itself is by estimated xGMI bytes
    # in plan/physical.py.
    max_bcast_bytes: int = 16 << 30
    # Emit a chrome-trace of the executed DAG (exec/tracing.py).
    trace: bool = False
    trace_path: str = "matrel_trace.json"
    # Deterministic mode: fixed reduction order (disables some overlap).
    deterministic: bool = False

    # --- optimizer switches (for tests / ablation) ---
    enable_transpose_pushdown: bool = True
    enable_chain_reorder: bool = True
    enable_cse: bool = True
    enable_scheme_selection: bool = True
    enable_scalar_folding: bool = True
    enable_agg_pushdown: bool = True

    @staticmethod
    def from_env() -> "MatrelConfig":
        cfg = MatrelConfig()
        cfg.block_size = _env_int("MATREL_BLOCK_SIZE", cfg.block_size)
        cfg.dtype = _env_str("MATREL_DTYPE", cfg.dtype)
        cfg.solve_dtype = _env_str("MATREL_SOLVE_DTYPE", cfg.solve_dtype)
        cfg.solve_max_side = _env_int("MATREL_SOLVE_MAX_SIDE",
                                      cfg.solve_max_side)
        cfg.gemm_precision = _env_str("MATREL_GEMM_PRECISION",
                                      cfg.gemm_precision)
        st = os.environ.get("MATREL_SPARSE_THRESHOLD")
        if st:
            cfg.sparse_threshold = float(st)
        sc = os.environ.get("MATREL_SPMM_DENSIFY_CROSSOVER")
        if sc:
            cfg.spmm_densify_crossover = float(sc)
        cfg.overlap_comm = _env_int("MATREL_OVERLAP_COMM",
                                    int(cfg.overlap_comm)) != 0
        cfg.summa_prefetch = _env_int("MATREL_SUMMA_PREFETCH",
                                      cfg.summa_prefetch)
        cfg.prep_stream = _env_int("MATREL_PREP_STREAM",
                                   int(cfg.prep_stream)) != 0
        cfg.max_bcast_bytes = _env_int("MATREL_MAX_BCAST_BYTES",
                                       cfg.max_bcast_bytes)
        cfg.deterministic = _env_int("MATREL_DETERMINISTIC",
                                     int(cfg.deterministic)) != 0
        cfg.trace = _env_int("MATREL_TRACE", int(cfg.trace)) != 0
        cfg.trace_path = _env_str("MATREL_TRACE_PATH", cfg.trace_path)
        g = os.environ.get("MATREL_GRID")
        if g:
            pr, pc = g.lower().split("x")
            cfg.grid = (int(pr), int(pc))
        return cfg


# The active config. matrel_amd.init() may replace it.
_config = MatrelConfig.from_env()


def get_config() -> MatrelConfig:
    return _config


def set_config(cfg: MatrelConfig) -> None:
    global _config
    _config = cfg
