"""Debug / race-detection aids.

Reference equivalent: none (the JVM memory model gives MatRel determinism
for free - SURVEY.md §5 "Race detection / sanitizers"); on ROCm the
equivalents are environment switches plus our deterministic-reduction
design:

  * serialize_kernels(): AMD_SERIALIZE_KERNEL=3 + HIP_LAUNCH_BLOCKING=1 -
    every kernel launch synchronous; turns async-race symptoms into
    immediate, attributable failures. Set BEFORE process start for full
    effect (env is read at HIP runtime init).
  * deterministic mode (MatrelConfig.deterministic): the engine's
    reductions are deterministic by construction - full/row/col reductions
    use fixed-order two-stage partials (csrc/reduce.hip: no float
    atomics), SUMMA accumulates k-panels in ascending k order on a single
    stream, and the bf16 GEMM accumulates in fp32 with a fixed K-tile
    order. The flag additionally disables comm/compute overlap so
    collective completion order cannot vary, and the sparse x sparse
    SpGEMM kernel (whose LDS-atomic accumulation is lane-order
    dependent) declines in favor of the fixed-order densified path
    (ops/blockops.spgemm_acc).
  * asan: build the extension with MATREL_ASAN=1 to add
    -fsanitize=address on the HOST side of the extension
    (python setup.py build_ext --inplace).
"""

from __future__ import annotations

import os


def serialize_kernels(enable: bool = True) -> None:
    if enable:
        os.environ["AMD_SERIALIZE_KERNEL"] = "3"
        os.environ["HIP_LAUNCH_BLOCKING"] = "1"
    else:
        os.environ.pop("AMD_SERIALIZE_KERNEL", None)
        os.environ.pop("HIP_LAUNCH_BLOCKING", None)


def enable_deterministic() -> None:
    from ..config import get_config
    cfg = get_config()
    cfg.deterministic = True
    cfg.overlap_comm = False


def debug_env_report() -> dict:
    keys = ("AMD_SERIALIZE_KERNEL", "HIP_LAUNCH_BLOCKING",
            "HSA_ENABLE_IPC_MODE_LEGACY", "PYTORCH_ROCM_ARCH",
            "MATREL_BLOCK_SIZE", "MATREL_DTYPE", "MATREL_GRID")
    return {k: os.environ.get(k) for k in keys}
