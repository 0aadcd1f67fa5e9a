"""matrel_amd: an MI355X-native distributed matrix-computation engine.

A from-scratch rebuild of purduedb/MatRel's capabilities (matrix-expression
DSL, cost-based matrix-plan optimizer, block-partitioned distributed
matrices, relational ops over matrix data) designed for one 8x MI355X node:
PyTorch-ROCm host, hand-written CDNA4 HIP MFMA kernels for the per-block
hot path, RCCL collectives over xGMI for SUMMA/broadcast distribution.
See SURVEY.md for the reference analysis this build follows.
"""

from .config import MatrelConfig, get_config, set_config
from .dsl import Matrix
from .session import MatrelSession
from .blocks.matrix import BlockMatrix
from .blocks import schemes
from .exec.runtime import get_runtime, init_runtime

__version__ = "0.1.0"

__all__ = [
    "MatrelConfig", "get_config", "set_config", "Matrix", "MatrelSession",
    "BlockMatrix", "schemes", "get_runtime", "init_runtime",
]
