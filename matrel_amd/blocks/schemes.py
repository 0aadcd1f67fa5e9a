"""Block-partition schemes: block -> owning rank maps.

The reference implements these as Spark Partitioner subclasses
(SURVEY.md §2.1 "Partitioners": RowPartitioner, ColumnPartitioner, grid /
block-cyclic). Here a scheme is a pure host-side function from block index
to owning rank on the logical (Pr x Pc) device grid - data movement it
implies is expressed separately as RCCL collectives / p2p by the executor.

Schemes:
  * RowScheme     - block-row i lives on one rank (row-cyclic over ranks).
  * ColScheme     - block-col j lives on one rank.
  * GridScheme    - 2D block-cyclic over the (Pr x Pc) grid: the SUMMA-native
                    layout (block (i,j) -> grid cell (i%Pr, j%Pc)).
  * BroadcastScheme - replicated on every rank (small operands).

All schemes are value-equal by (kind, params) so the optimizer can compare
the scheme of a plan node with the scheme an operator wants.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple


@dataclass(frozen=True)
class Scheme:
    kind: str = "grid"

    def owner(self, bi: int, bj: int, world: int, grid: Tuple[int, int]) -> int:
        raise NotImplementedError

    def is_replicated(self) -> bool:
        return False

    def transpose(self) -> "Scheme":
        """Scheme of A.T's blocks given A's scheme (block (i,j)->(j,i))."""
        return self


@dataclass(frozen=True)
class RowScheme(Scheme):
    kind: str = "row"

    def owner(self, bi: int, bj: int, world: int, grid: Tuple[int, int]) -> int:
        return bi % world

    def transpose(self) -> "Scheme":
        return ColScheme()


@dataclass(frozen=True)
class ColScheme(Scheme):
    kind: str = "col"

    def owner(self, bi: int, bj: int, world: int, grid: Tuple[int, int]) -> int:
        return bj % world

    def transpose(self) -> "Scheme":
        return RowScheme()


@dataclass(frozen=True)
class GridScheme(Scheme):
    kind: str = "grid"

    def owner(self, bi: int, bj: int, world: int, grid: Tuple[int, int]) -> int:
        pr, pc = grid
        return (bi % pr) * pc + (bj % pc)

    def transpose(self) -> "Scheme":
        # Transposed blocks land on the transposed grid cell; for a
        # non-square grid that is a different rank map, which the executor
        # realizes by an actual repartition when needed. We keep GridScheme
        # as the nominal result (square grids: exact).
        return GridScheme()


@dataclass(frozen=True)
class BroadcastScheme(Scheme):
    kind: str = "broadcast"

    def owner(self, bi: int, bj: int, world: int, grid: Tuple[int, int]) -> int:
        return -1  # every rank

    def is_replicated(self) -> bool:
        return True


ROW = RowScheme()
COL = ColScheme()
GRID = GridScheme()
BCAST = BroadcastScheme()


def scheme_from_name(name: str) -> Scheme:
    return {"row": ROW, "col": COL, "grid": GRID, "broadcast": BCAST}[name]
