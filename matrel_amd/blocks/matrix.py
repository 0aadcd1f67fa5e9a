"""Block-partitioned distributed matrix.

Reference equivalent: MatRel's RDD[((rowBlk, colBlk), MLMatrix)] with
~1000x1000 blocks (SURVEY.md §2.1 "Block-matrix representation"). The
MI355X-native design differs deliberately:

  * Blocks are LARGE (default 16384^2: bf16 block = 512 MiB) - one MI355X
    holds 288 GB of HBM3E, and MFMA GEMM efficiency wants >= 8k tiles.
    Block size is per-matrix, so block-sparse matrices can use finer grids.
  * A block is a torch.Tensor resident on this rank's GPU (or CPU in test
    mode). ABSENT blocks are implicit zeros - this is the block-sparse
    representation (BASELINE.json config 4) and also what block-granular
    selection pruning produces.
  * Ownership is a pure function of (block index, scheme, device grid) -
    there is no shuffle machinery; movement is explicit RCCL collectives
    issued by the executor.

Supports ragged edge blocks (m, n need not divide block_size).
"""

from __future__ import annotations

import math
import weakref
from typing import Dict, Iterator, Optional, Tuple

import torch
import torch.distributed as dist

from ..config import get_config
from ..exec.runtime import get_runtime
from .schemes import GRID, Scheme


def _densify(t: torch.Tensor) -> torch.Tensor:
    return t.to_dense() if t.layout == torch.sparse_csr else t

BlockIndex = Tuple[int, int]


class _BlockStore(dict):
    """Per-matrix block dict that invalidates the matrix's cached global
    stats (presence / nnz maps) and derived-panel caches on ANY mutation.

    Much of the engine writes `m.blocks[idx] = t` directly; before this
    existed only `set_block` invalidated, so a stale nnz map could make the
    SUMMA CSR-wire encoder size broadcast buffers from outdated values
    (ADVICE.md round 1). Now every mutation path is safe.

    The owner backref is a WEAKREF: matrix -> .blocks -> owner as a
    strong ref would be a reference cycle, and multi-hundred-GB result
    matrices would then die only at gc intervals instead of by refcount
    (caught as an OOM regression on the near-capacity 163840^2 bench).
    """

    __slots__ = ("_owner",)

    def __init__(self, owner: "BlockMatrix", *args) -> None:
        super().__init__(*args)
        self._owner = weakref.ref(owner)

    def _inv(self) -> None:
        o = self._owner()
        if o is not None:
            o._invalidate_stats()

    def __setitem__(self, k, v) -> None:
        super().__setitem__(k, v)
        self._inv()

    def __delitem__(self, k) -> None:
        super().__delitem__(k)
        self._inv()

    def pop(self, *args):
        r = super().pop(*args)
        self._inv()
        return r

    def popitem(self):
        r = super().popitem()
        self._inv()
        return r

    def update(self, *args, **kw) -> None:
        super().update(*args, **kw)
        self._inv()

    def clear(self) -> None:
        super().clear()
        self._inv()

    def setdefault(self, k, default=None):
        r = super().setdefault(k, default)
        self._inv()
        return r


def _torch_dtype(name: str) -> torch.dtype:
    return {
        "bf16": torch.bfloat16,
        "fp16": torch.float16,
        "fp32": torch.float32,
        "fp64": torch.float64,
    }[name]


class BlockMatrix:
    """A (possibly distributed) block-partitioned matrix."""

    def __init__(self, nrows: int, ncols: int, block_size: Optional[int] = None,
                 scheme: Scheme = GRID, dtype: Optional[torch.dtype] = None,
                 device: Optional[torch.device] = None) -> None:
        cfg = get_config()
        rt = get_runtime()
        self.nrows = int(nrows)
        self.ncols = int(ncols)
        bs = int(block_size or cfg.block_size)
        if bs <= 0:
            # auto block size: single-process dense work wants ONE block per
            # matrix (the raw MFMA GEMM, no block orchestration overhead) up
            # to 65536 (8 GiB bf16 - comfortable in 288 GB HBM3E); multi-GPU
            # keeps 16384 so SUMMA has k-panel granularity to overlap
            # broadcasts with compute.
            if rt.world_size == 1:
                bs = max(1, min(65536, max(int(nrows), int(ncols))))
            else:
                bs = 16384
        self.block_size = bs
        self.scheme = scheme
        self.device = device if device is not None else rt.device
        if dtype is None:
            dtype = _torch_dtype(cfg.dtype) if self.device.type == "cuda" \
                else torch.float32
        self.dtype = dtype
        # Local (this-rank) blocks only. _BlockStore invalidates the
        # cached presence/nnz maps on every mutation.
        self._presence = None
        self._nnzmap = None
        self.blocks: Dict[BlockIndex, torch.Tensor] = _BlockStore(self)

    def _invalidate_stats(self) -> None:
        self._presence = None
        self._nnzmap = None
        # derived-panel caches keyed by block index are stale too
        for attr in ("_bt_cache", "_bt_strip_cache", "_mx8_cache"):
            if getattr(self, attr, None):
                setattr(self, attr, {})

    # --- geometry -------------------------------------------------------

    @property
    def grid_rows(self) -> int:
        return max(1, math.ceil(self.nrows / self.block_size))

    @property
    def grid_cols(self) -> int:
        return max(1, math.ceil(self.ncols / self.block_size))

    @property
    def shape(self) -> Tuple[int, int]:
        return (self.nrows, self.ncols)

    def block_shape(self, bi: int, bj: int) -> Tuple[int, int]:
        bs = self.block_size
        h = min(bs, self.nrows - bi * bs)
        w = min(bs, self.ncols - bj * bs)
        return (h, w)

    def block_indices(self) -> Iterator[BlockIndex]:
        for bi in range(self.grid_rows):
            for bj in range(self.grid_cols):
                yield (bi, bj)

    # --- ownership ------------------------------------------------------

    def owner(self, bi: int, bj: int) -> int:
        rt = get_runtime()
        return self.scheme.owner(bi, bj, rt.world_size, rt.grid)

    def is_local(self, bi: int, bj: int) -> bool:
        rt = get_runtime()
        o = self.owner(bi, bj)
        return o == -1 or o == rt.rank

    def owned_indices(self) -> Iterator[BlockIndex]:
        for idx in self.block_indices():
            if self.is_local(*idx):
                yield idx

    # --- block access ---------------------------------------------------

    def get_block(self, bi: int, bj: int) -> Optional[torch.Tensor]:
        """Local block or None if absent (= zero) or remote."""
        return self.blocks.get((bi, bj))

    def get_block_dense(self, bi: int, bj: int) -> torch.Tensor:
        """Local block as a DENSE tensor (zeros for absent blocks, CSR
        blocks densified)."""
        t = self.blocks.get((bi, bj))
        if t is None:
            t = torch.zeros(self.block_shape(bi, bj), dtype=self.dtype,
                            device=self.device)
        return _densify(t)

    def set_block(self, bi: int, bj: int, t: torch.Tensor) -> None:
        assert t.shape == self.block_shape(bi, bj), \
            f"block {(bi, bj)}: {tuple(t.shape)} != {self.block_shape(bi, bj)}"
        self.blocks[(bi, bj)] = t.to(device=self.device, dtype=self.dtype)

    def coll_device(self) -> torch.device:
        """Device for small bookkeeping collectives: RCCL needs GPU
        tensors, gloo needs CPU tensors."""
        rt = get_runtime()
        return self.device if rt.backend == "nccl" else torch.device("cpu")

    def presence(self) -> torch.Tensor:
        """Global (grid_rows x grid_cols) uint8 map of present blocks
        (one all_reduce; cached). Every rank agrees on the result, which
        lets repartition/transpose/SUMMA skip absent blocks without
        per-block flag exchanges."""
        cached = getattr(self, "_presence", None)
        if cached is not None:
            return cached
        rt = get_runtime()
        pm = torch.zeros(self.grid_rows, self.grid_cols, dtype=torch.uint8)
        for (bi, bj) in self.blocks:
            pm[bi, bj] = 1
        if rt.is_distributed and not self.scheme.is_replicated():
            dev_pm = pm.to(self.coll_device())
            dist.all_reduce(dev_pm, op=dist.ReduceOp.MAX)
            pm = dev_pm.cpu()
        self._presence = pm
        return pm

    def local_items(self) -> Iterator[Tuple[BlockIndex, torch.Tensor]]:
        return iter(self.blocks.items())

    def nnz_map(self) -> torch.Tensor:
        """Global (grid_rows x grid_cols) int64 map of STORED nnz per
        block: CSR blocks report their nnz, dense blocks report h*w
        (no scan - storage state is the owner's sparsify decision), absent
        blocks 0. One all_reduce, cached; every rank agrees, which lets
        SUMMA decide dense-vs-CSR wire format group-uniformly and size
        the variable-length CSR broadcast buffers without a handshake."""
        cached = getattr(self, "_nnzmap", None)
        if cached is not None:
            return cached
        rt = get_runtime()
        nm = torch.zeros(self.grid_rows, self.grid_cols, dtype=torch.int64)
        for (bi, bj), t in self.blocks.items():
            if t.layout == torch.sparse_csr:
                nm[bi, bj] = t._nnz()
            else:
                nm[bi, bj] = t.shape[0] * t.shape[1]
        if rt.is_distributed and not self.scheme.is_replicated():
            dev_nm = nm.to(self.coll_device())
            dist.all_reduce(dev_nm, op=dist.ReduceOp.SUM)
            nm = dev_nm.cpu()
        self._nnzmap = nm
        return nm

    # --- stats (feed the optimizer's sparsity-aware cost model) ---------

    def block_density(self) -> float:
        """Fraction of blocks present, across all ranks."""
        total = self.grid_rows * self.grid_cols
        return float(self.presence().sum().item()) / total

    # --- constructors ---------------------------------------------------

    @classmethod
    def zeros(cls, m: int, n: int, **kw) -> "BlockMatrix":
        return cls(m, n, **kw)

    @classmethod
    def _filled(cls, m: int, n: int, fill, seed: Optional[int], **kw
                ) -> "BlockMatrix":
        out = cls(m, n, **kw)
        # generate directly on the target device (a 16384^2 block takes
        # ~ms on GPU vs seconds through host randn + H2D); per-block
        # seeding keeps the fill reproducible per (seed, bi, bj) within a
        # device type.
        gen_dev = "cuda" if out.device.type == "cuda" else "cpu"
        for (bi, bj) in out.owned_indices():
            shape = out.block_shape(bi, bj)
            if seed is not None:
                g = torch.Generator(device=gen_dev)
                g.manual_seed((seed * 1000003 + bi * 131 + bj) & 0x7FFFFFFF)
                t = fill(shape, g).to(device=out.device, dtype=out.dtype)
            else:
                t = fill(shape, None).to(device=out.device, dtype=out.dtype)
            out.blocks[(bi, bj)] = t
        return out

    @classmethod
    def randn(cls, m: int, n: int, seed: Optional[int] = 0, **kw
              ) -> "BlockMatrix":
        """Block-reproducible N(0,1) fill: block (i,j) content depends only
        on (seed, i, j), so the same logical matrix materializes under any
        scheme / world size (used by tests and the bench)."""
        return cls._filled(
            m, n,
            lambda s, g: torch.randn(*s, generator=g,
                                     device=g.device if g else None),
            seed, **kw)

    @classmethod
    def rand(cls, m: int, n: int, seed: Optional[int] = 0, **kw
             ) -> "BlockMatrix":
        return cls._filled(
            m, n,
            lambda s, g: torch.rand(*s, generator=g,
                                    device=g.device if g else None),
            seed, **kw)

    @classmethod
    def from_dense(cls, t: torch.Tensor, sparsify: bool = True,
                   **kw) -> "BlockMatrix":
        """Split a full tensor (available on every rank) into blocks.
        Blocks below the config sparse_threshold are stored CSR
        (reference: auto dense/sparse per block by nnz)."""
        assert t.dim() == 2
        from ..ops.blockops import maybe_sparsify
        thr = get_config().sparse_threshold if sparsify else 0.0
        out = cls(t.shape[0], t.shape[1], **kw)
        bs = out.block_size
        for (bi, bj) in out.owned_indices():
            blk = t[bi * bs:(bi + 1) * bs, bj * bs:(bj + 1) * bs]
            if blk.count_nonzero() == 0:
                continue  # keep zero blocks implicit
            dense = blk.contiguous().to(device=out.device, dtype=out.dtype)
            out.blocks[(bi, bj)] = maybe_sparsify(dense, thr)
        return out

    @classmethod
    def sparse_random(cls, m: int, n: int, density: float, seed: int = 0,
                      **kw) -> "BlockMatrix":
        """Element-level sparse random matrix: every block stored CSR with
        ~density fraction of nonzeros, reproducible by block index."""
        out = cls(m, n, **kw)
        for (bi, bj) in out.owned_indices():
            g = torch.Generator(device="cpu")
            g.manual_seed((seed * 999983 + bi * 131 + bj) & 0x7FFFFFFF)
            shape = out.block_shape(bi, bj)
            mask = torch.rand(*shape, generator=g) < density
            vals = torch.randn(*shape, generator=g) * mask
            t = vals.to(device=out.device, dtype=out.dtype)
            out.blocks[(bi, bj)] = t.to_sparse_csr()
        return out

    @classmethod
    def block_sparse_random(cls, m: int, n: int, block_density: float,
                            seed: int = 0, **kw) -> "BlockMatrix":
        """Random block-sparse matrix: each block present with probability
        block_density, reproducibly by block index (config 4)."""
        out = cls(m, n, **kw)
        for (bi, bj) in out.owned_indices():
            g = torch.Generator(device="cpu")
            g.manual_seed((seed * 1000003 + bi * 131 + bj) & 0x7FFFFFFF)
            if torch.rand(1, generator=g).item() >= block_density:
                continue
            t = torch.randn(*out.block_shape(bi, bj), generator=g)
            out.blocks[(bi, bj)] = t.to(device=out.device, dtype=out.dtype)
        return out

    # --- materialization (small matrices / tests only) ------------------

    def to_dense(self) -> torch.Tensor:
        """Assemble the full matrix on every rank (collective when
        distributed). Only for small results and tests."""
        rt = get_runtime()
        full = torch.zeros(self.nrows, self.ncols, dtype=torch.float32,
                           device=self.device)
        bs = self.block_size
        for (bi, bj) in self.block_indices():
            owner = self.owner(bi, bj)
            if owner == -1 or not rt.is_distributed:
                t = self.blocks.get((bi, bj))
                if t is not None:
                    t = _densify(t)
                    full[bi * bs:bi * bs + t.shape[0],
                         bj * bs:bj * bs + t.shape[1]] = t.float()
            else:
                shape = self.block_shape(bi, bj)
                if rt.rank == owner:
                    # contiguous is mandatory: gloo/RCCL collectives on a
                    # strided view (e.g. a ragged c[:h, :w] crop) send the
                    # wrong bytes silently
                    t = self.get_block_dense(bi, bj).float().contiguous()
                else:
                    t = torch.zeros(shape, dtype=torch.float32,
                                    device=self.device)
                dist.broadcast(t, src=owner)
                full[bi * bs:bi * bs + shape[0],
                     bj * bs:bj * bs + shape[1]] = t
        return full

    # --- scheme movement ------------------------------------------------

    def repartition(self, scheme: Scheme) -> "BlockMatrix":
        """Move blocks to the ownership map of `scheme`.

        Reference: RDD repartition/shuffle. Here: per-block broadcast from
        the current owner (xGMI p2p under RCCL). Replication targets
        (BroadcastScheme) receive every block everywhere.
        """
        rt = get_runtime()
        if scheme == self.scheme:
            return self
        out = BlockMatrix(self.nrows, self.ncols, self.block_size, scheme,
                          self.dtype, self.device)
        if not rt.is_distributed:
            out.blocks.update(self.blocks)
            return out
        from ..exec.metrics import add_comm
        pm = self.presence()
        # CSR wire (round 2): element-sparse blocks move as their CSR
        # arrays - nnz-proportional xGMI bytes, and receivers KEEP the
        # sparse storage (previously every move densified). The decision
        # is group-uniform: derived from the globally-synced nnz map with
        # the same predicate the SUMMA wire uses (only CSR-stored blocks
        # can satisfy it - dense-stored blocks report h*w).
        nm = self.nnz_map()
        thr = get_config().sparse_threshold

        def wire_sparse(bi: int, bj: int) -> bool:
            h, w = self.block_shape(bi, bj)
            return int(nm[bi, bj]) < thr * h * w

        def csr_bufs(bi: int, bj: int):
            """(crow, cols, vals) buffers; source fills, others allocate
            (sizes agreed from the nnz map + geometry)."""
            h, w = self.block_shape(bi, bj)
            nz = int(nm[bi, bj])
            if rt.rank == self.owner(bi, bj):
                t = self.blocks[(bi, bj)]
                if t.layout != torch.sparse_csr:
                    t = t.to_sparse_csr()
                return (t.crow_indices().to(torch.int32).contiguous(),
                        t.col_indices().to(torch.int32).contiguous(),
                        t.values().contiguous())
            return (torch.empty(h + 1, dtype=torch.int32,
                                device=self.device),
                    torch.empty(nz, dtype=torch.int32, device=self.device),
                    torch.empty(nz, dtype=self.dtype, device=self.device))

        for (bi, bj) in self.block_indices():
            src = self.owner(bi, bj)
            dst = scheme.owner(bi, bj, rt.world_size, rt.grid)
            if src == -1:
                # already replicated: every rank has it locally (or zero)
                if dst == -1 or dst == rt.rank:
                    t = self.blocks.get((bi, bj))
                    if t is not None:
                        out.blocks[(bi, bj)] = t
                continue
            if not pm[bi, bj]:
                continue
            sp = wire_sparse(bi, bj)
            if dst == -1:
                if sp:
                    crow, cols, vals = csr_bufs(bi, bj)
                    for b in (crow, cols, vals):
                        dist.broadcast(b, src=src)
                    add_comm("repartition_bcast_csr",
                             crow.numel() * 4 + cols.numel() * 4
                             + vals.numel() * vals.element_size())
                    out.blocks[(bi, bj)] = torch.sparse_csr_tensor(
                        crow, cols, vals, size=self.block_shape(bi, bj))
                    continue
                t = (_densify(self.blocks[(bi, bj)]).contiguous()
                     if rt.rank == src else
                     torch.empty(self.block_shape(bi, bj), dtype=self.dtype,
                                 device=self.device))
                dist.broadcast(t, src=src)
                add_comm("repartition_bcast",
                         t.numel() * t.element_size())
                out.blocks[(bi, bj)] = t
            elif src == dst:
                if rt.rank == src:
                    out.blocks[(bi, bj)] = self.blocks[(bi, bj)]
            elif sp:
                if rt.rank == src:
                    crow, cols, vals = csr_bufs(bi, bj)
                    for b in (crow, cols, vals):
                        dist.send(b, dst=dst)
                    add_comm("repartition_send_csr",
                             crow.numel() * 4 + cols.numel() * 4
                             + vals.numel() * vals.element_size())
                elif rt.rank == dst:
                    crow, cols, vals = csr_bufs(bi, bj)
                    for b in (crow, cols, vals):
                        dist.recv(b, src=src)
                    out.blocks[(bi, bj)] = torch.sparse_csr_tensor(
                        crow, cols, vals, size=self.block_shape(bi, bj))
            else:
                if rt.rank == src:
                    tt = _densify(self.blocks[(bi, bj)]).contiguous()
                    dist.send(tt, dst=dst)
                    add_comm("repartition_send",
                             tt.numel() * tt.element_size())
                elif rt.rank == dst:
                    t = torch.empty(self.block_shape(bi, bj), dtype=self.dtype,
                                    device=self.device)
                    dist.recv(t, src=src)
                    out.blocks[(bi, bj)] = t
        return out

    def __repr__(self) -> str:
        return (f"BlockMatrix({self.nrows}x{self.ncols}, bs={self.block_size},"
                f" grid={self.grid_rows}x{self.grid_cols},"
                f" scheme={self.scheme.kind}, dtype={self.dtype},"
                f" local_blocks={len(self.blocks)})")
