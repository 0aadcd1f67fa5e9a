"""Distributed matmul algorithms: SUMMA, broadcast-MM, local.

Reference equivalent: MatRel's physical matmul strategies:
broadcast-MM (small side replicated), CPMM (shuffle+reduceByKey on the inner
index) and RMM (replication) - the latter two are replaced wholesale by
SUMMA, the xGMI-idiomatic algorithm: per k-step, the owners
broadcast an A block-panel along each grid row and a B block-panel along
each grid column (RCCL broadcast on row/col sub-communicators), and every
rank accumulates its C blocks with the local MFMA GEMM.

Comm/compute overlap: panel broadcasts for step k+1 are issued async
(RCCL's internal stream) before the step-k GEMMs are launched on the
compute stream - double-buffered panels, no barrier between steps.

All matrices here are in GridScheme (2D block-cyclic); the executor
repartitions operands first when the planner chose SUMMA.
"""

from __future__ import annotations

from typing import Dict, Tuple

import torch
import torch.distributed as dist

from ..blocks.matrix import BlockMatrix
from ..config import get_config
from ..ops import blockops
from .metrics import add_comm
from .runtime import get_runtime


def _presence_map(m: BlockMatrix) -> torch.Tensor:
    return m.presence()


_prep_stream: torch.cuda.Stream = None


def _get_prep_stream():
    """Dedicated HIP stream for owner-side panel preparation
    (transposes, pads, mx8 conversions, CSR wire packing): prep work for
    step k+1 runs UNDER the step-k GEMMs already enqueued on the compute
    stream instead of serializing between their launches (SURVEY.md §2.2
    "host DAG executor over HIP streams"; ROADMAP r1 item 3)."""
    global _prep_stream
    if _prep_stream is None:
        _prep_stream = torch.cuda.Stream()
    return _prep_stream


def _cache_has_headroom(nbytes: int) -> bool:
    """Only grow a convenience cache while the device keeps at least 4x
    the entry size PLUS 16 GiB free - caches are a speedup, not worth an
    OOM on near-capacity jobs (288 GB fills up fast at 65536^2 blocks)."""
    try:
        free, _total = torch.cuda.mem_get_info()
    except Exception:
        return True
    return free > 4 * nbytes + (16 << 30)


def _bt_cached(b: BlockMatrix, k: int, bj: int,
               bbd: torch.Tensor) -> torch.Tensor:
    """Transposed-block cache on the B matrix (GPU only): iterative
    workloads (bench steps, power iterations) reuse B across calls, so
    the transpose is paid once per block ever, not once per multiply.
    Under memory pressure the transpose is recomputed instead of cached."""
    if not bbd.is_cuda:
        return bbd.t()
    cache = getattr(b, "_bt_cache", None)
    if cache is None:
        cache = b._bt_cache = {}
    bt = cache.get((k, bj))
    if bt is None:
        bt = blockops.transpose(bbd)
        if _cache_has_headroom(bt.numel() * bt.element_size()):
            cache[(k, bj)] = bt
    return bt


_MXQ_PRECS = ("mx8", "mx6", "mx4")


def _mxq_convert(prec: str, t: torch.Tensor):
    if prec == "mx4":
        return blockops.to_mx4(t)
    if prec == "mx6":
        return blockops.to_mx6(t)
    return blockops.to_mx8(t)


def _mxq_gemm(prec: str, c, a_p, a_sc, bt_p, bt_sc, beta0: bool) -> None:
    if prec == "mx4":
        blockops.gemm_bt_mx4_acc(c, a_p, a_sc, bt_p, bt_sc, beta0=beta0)
    elif prec == "mx6":
        blockops.gemm_bt_mx6_acc(c, a_p, a_sc, bt_p, bt_sc, beta0=beta0)
    else:
        blockops.gemm_bt_mx8_acc(c, a_p, a_sc, bt_p, bt_sc, beta0=beta0)


def _mxq_pbytes(prec, bs: int) -> int:
    """Quantized payload bytes per row of a bs-wide panel."""
    if prec == "mx4":
        return bs // 2
    if prec == "mx6":
        return bs * 3 // 4
    return bs


def _mx8_cached(m: BlockMatrix, key, tensor_bf16: torch.Tensor,
                prec: str = "mx8"):
    """Per-matrix cache of MX-fp8/fp4 conversions (payloads are 1/2 or
    1/4 the bf16 bytes; reused across every multiply touching the
    block)."""
    cache = getattr(m, "_mx8_cache", None)
    if cache is None:
        cache = m._mx8_cache = {}
    key = (prec,) + (key if isinstance(key, tuple) else (key,))
    ent = cache.get(key)
    if ent is None:
        ent = _mxq_convert(prec, tensor_bf16)
        if _cache_has_headroom(sum(t.numel() * t.element_size()
                                   for t in ent)):
            cache[key] = ent
    return ent


def _mx8_usable(t: torch.Tensor, prec: str = "mx8") -> bool:
    # mx8 has a legacy byte-scale layout + BK=64 kernel for cols % 32;
    # the packed-scale mx6/mx4 converters and their k128 kernels need
    # cols % 128 (a ragged edge of width 64 mod 128 passes the mx8 rule
    # but raises in mx6_convert - caught by the refinement example).
    kmod = 32 if prec == "mx8" else 128
    return (t.is_cuda and t.dtype == torch.bfloat16
            and t.shape[0] % 256 == 0 and t.shape[1] % kmod == 0)


def _try_batch_local(a: BlockMatrix, b: BlockMatrix,
                     out: BlockMatrix) -> bool:
    """Fine-block fast path: hand the whole per-block GEMM task list to
    the native C++ batch executor (csrc/ext.cpp gemm_bt_batch) - one
    Python->native transition for the entire plan instead of one per
    block pair. Used for tile-aligned all-dense bf16 plans with many
    small blocks (block-sparse workloads); falls back otherwise."""
    if out.device.type != "cuda" or out.dtype != torch.bfloat16:
        return False
    if get_config().gemm_precision in _MXQ_PRECS:
        return False
    bs = out.block_size
    if bs % 256 != 0 or bs > 8192:
        return False
    if (a.nrows % bs or a.ncols % bs or b.ncols % bs
            or a.block_size != bs or b.block_size != bs):
        return False
    for _, t in a.local_items():
        if blockops.is_sparse(t) or t.dtype != torch.bfloat16:
            return False
    for _, t in b.local_items():
        if blockops.is_sparse(t) or t.dtype != torch.bfloat16:
            return False
    accs, acc_keys = [], {}
    a_list, a_idx = [], {}
    bt_list, bt_idx = [], {}
    tasks = []
    for (bi, bj) in out.owned_indices():
        ks = [k for k in range(a.grid_cols)
              if (bi, k) in a.blocks and (k, bj) in b.blocks]
        if not ks:
            continue
        ai = acc_keys.get((bi, bj))
        if ai is None:
            ai = len(accs)
            acc_keys[(bi, bj)] = ai
            accs.append(torch.empty(out.block_shape(bi, bj),
                                    dtype=torch.float32,
                                    device=out.device))
        for pos, k in enumerate(ks):
            if (bi, k) not in a_idx:
                a_idx[(bi, k)] = len(a_list)
                a_list.append(a.blocks[(bi, k)].contiguous())
            if (k, bj) not in bt_idx:
                bt_idx[(k, bj)] = len(bt_list)
                bt_list.append(_bt_cached(b, k, bj, b.blocks[(k, bj)]))
            tasks.append((ai, a_idx[(bi, k)], bt_idx[(k, bj)],
                          1 if pos == 0 else 0))
    if len(tasks) < 16:
        return False
    from ..ops.backend import require_ext
    require_ext().gemm_bt_batch(
        accs, a_list, bt_list,
        torch.tensor(tasks, dtype=torch.int64))
    for (bi, bj), ai in acc_keys.items():
        out.blocks[(bi, bj)] = accs[ai].to(out.dtype)
    return True


def _pad_t(t: torch.Tensor, bs: int) -> torch.Tensor:
    if t.shape == (bs, bs):
        return t
    pad = torch.zeros(bs, bs, dtype=t.dtype, device=t.device)
    pad[: t.shape[0], : t.shape[1]] = t
    return pad


def _mx8_padded(m: BlockMatrix, key, t: torch.Tensor, bs: int,
                prec: str = "mx8"):
    """Cached MX-fp8/fp4 conversion of a (zero-padded) block."""
    cache = getattr(m, "_mx8_cache", None)
    if cache is None:
        cache = m._mx8_cache = {}
    key = (prec,) + (key if isinstance(key, tuple) else (key,))
    ent = cache.get(key)
    if ent is None:
        ent = _mxq_convert(prec, _pad_t(blockops.densify(t), bs))
        if _cache_has_headroom(sum(x.numel() * x.element_size()
                                   for x in ent)):
            cache[key] = ent
    return ent


def _bt_strip(b: BlockMatrix, k: int) -> torch.Tensor:
    """Concatenated transposed B block-row k: [N_total, bs] bf16 (cached
    on the matrix; absent blocks materialize as zeros)."""
    cache = getattr(b, "_bt_strip_cache", None)
    if cache is None:
        cache = b._bt_strip_cache = {}
    s = cache.get(k)
    if s is None:
        parts = [_bt_cached(b, k, bj, blockops.densify(
                     b.get_block_dense(k, bj)))
                 for bj in range(b.grid_cols)]
        s = torch.cat(parts, dim=0).contiguous()
        if not s.is_cuda or _cache_has_headroom(
                s.numel() * s.element_size()):
            cache[k] = s
    return s


def _try_strip_local(a: BlockMatrix, b: BlockMatrix,
                     out: BlockMatrix) -> bool:
    """Sparse-A strip mode: fuse a present A block's whole row of
    per-block GEMMs into ONE wide-N GEMM against the concatenated
    (transposed) B block-row - C(bi, :) += A(bi, k) @ B(k, :). Fine
    blocks (config 4's 1%-block-density SpMM) then run in the
    large-grid phased-GEMM regime instead of many under-filled small
    launches."""
    if out.device.type != "cuda" or out.dtype != torch.bfloat16:
        return False
    if get_config().gemm_precision in _MXQ_PRECS:
        return False
    bs = out.block_size
    if bs % 256 != 0 or bs > 4096 or b.grid_cols < 4:
        return False
    if (a.nrows % bs or a.ncols % bs or b.ncols % bs
            or a.block_size != bs or b.block_size != bs):
        return False
    # only pays off when A is block-sparse (dense A prefers the fused
    # per-block path). Use the GLOBAL presence fraction - the local block
    # count under a distributed scheme would make dense matrices look
    # sparse (1/world of the blocks are local).
    if a.block_density() > 0.5:
        return False
    for _, t in a.local_items():
        if blockops.is_sparse(t) or t.dtype != torch.bfloat16:
            return False
    n_total = b.ncols
    my_bis = sorted({bi for (bi, k) in a.blocks})
    strip_bytes = 4 * bs * n_total
    if strip_bytes * len(my_bis) > 64 << 30:
        return False
    # Split rows by contribution count. At 1% block density most rows
    # have exactly ONE contributing A block: those run the fused
    # beta0+bf16-out GEMM straight into a bf16 strip (no fp32 strip, no
    # conversion pass - the rocprof profile showed the f32->bf16 slice
    # conversions costing ~20% of config-4 time). Multi-k rows keep the
    # fp32-accumulating native batch executor.
    row_ks = {bi: sorted(k for (i, k) in a.blocks if i == bi)
              for bi in my_bis}
    single = [bi for bi in my_bis if len(row_ks[bi]) == 1]
    multi = [bi for bi in my_bis if len(row_ks[bi]) > 1]

    strips: Dict[int, torch.Tensor] = {}
    dummy = None
    for bi in single:
        k = row_ks[bi][0]
        ab = a.blocks[(bi, k)].contiguous()
        bt = _bt_strip(b, k)
        if dummy is None:
            dummy = torch.empty(bs, n_total, dtype=torch.float32,
                                device=out.device)  # never touched (beta0)
        ob = torch.empty(bs, n_total, dtype=out.dtype, device=out.device)
        if blockops.gemm_bt_final(dummy, ob, ab, bt, beta0=True):
            strips[bi] = ob
        else:
            multi.append(bi)  # fused geometry unsupported: f32 path

    accs, acc_idx = [], {}
    a_list, a_map = [], {}
    bt_list, bt_map = [], {}
    tasks = []
    for bi in multi:
        ks = row_ks[bi]
        ai = len(accs)
        acc_idx[bi] = ai
        accs.append(torch.empty(bs, n_total, dtype=torch.float32,
                                device=out.device))
        for pos, k in enumerate(ks):
            if (bi, k) not in a_map:
                a_map[(bi, k)] = len(a_list)
                a_list.append(a.blocks[(bi, k)].contiguous())
            if k not in bt_map:
                bt_map[k] = len(bt_list)
                bt_list.append(_bt_strip(b, k))
            tasks.append((ai, a_map[(bi, k)], bt_map[k],
                          1 if pos == 0 else 0))
    if tasks:
        from ..ops.backend import require_ext
        require_ext().gemm_bt_batch(accs, a_list, bt_list,
                                    torch.tensor(tasks, dtype=torch.int64))
    for bi in my_bis:
        fstrip = strips.get(bi)
        strip = fstrip if fstrip is not None else (
            accs[acc_idx[bi]] if bi in acc_idx else None)
        if strip is None:
            continue
        for bj in range(out.grid_cols):
            if not out.is_local(bi, bj):
                continue
            sl = strip[:, bj * bs:(bj + 1) * bs]
            out.blocks[(bi, bj)] = (sl.contiguous() if fstrip is not None
                                    else sl.to(out.dtype).contiguous())
    return True


def _csr_density(t: torch.Tensor) -> float:
    return t._nnz() / max(1, t.shape[0] * t.shape[1])


def _route_dense(t: torch.Tensor, device: torch.device) -> bool:
    """True when a CSR block computes faster densified on the MFMA pipe
    than through the CSR kernels (measured crossover ~1% density at
    8-16k blocks; profiles/ spmm isolation entry). Storage and wire stay
    CSR - this only routes the local product."""
    return (device.type == "cuda" and blockops.is_sparse(t)
            and _csr_density(t) >= get_config().spmm_densify_crossover)


def local_matmul(a: BlockMatrix, b: BlockMatrix, out: BlockMatrix) -> None:
    """Pure-local block multiply: every needed A/B block is already on the
    rank that owns the corresponding C block (single process, or
    partition-aware alignment: A row-partitioned + B replicated, etc.).
    Loops (j, k) outer so each B block is transposed once for the B^T MFMA
    kernel and reused across the block-row loop."""
    if _try_strip_local(a, b, out):
        return
    if _try_batch_local(a, b, out):
        return
    fused_ok = (out.device.type == "cuda" and out.dtype == torch.bfloat16)
    _prec = get_config().gemm_precision
    mxq = _prec if _prec in _MXQ_PRECS else None
    mx8 = mxq is not None
    a_dense_cache: Dict[Tuple[int, int], torch.Tensor] = {}

    def a_routed(bi: int, k: int) -> torch.Tensor:
        t = a.blocks[(bi, k)]
        if _route_dense(t, out.device):
            d = a_dense_cache.get((bi, k))
            if d is None:
                d = blockops.densify(t)
                a_dense_cache[(bi, k)] = d
            return d
        return t

    for (bi, bj) in out.owned_indices():
        ks = [k for k in range(a.grid_cols)
              if (bi, k) in a.blocks and (k, bj) in b.blocks]
        if not ks:
            continue
        shape = out.block_shape(bi, bj)
        c = torch.empty(shape, dtype=torch.float32, device=out.device)
        any_sparse = any(blockops.is_sparse(a_routed(bi, k)) for k in ks)
        use_fused = fused_ok and not any_sparse
        outblk = (torch.empty(shape, dtype=out.dtype, device=out.device)
                  if use_fused else None)
        fused_done = False
        for idx, k in enumerate(ks):
            ab = a_routed(bi, k)
            bb = b.blocks[(k, bj)]
            first = idx == 0
            last = idx == len(ks) - 1
            if blockops.is_sparse(ab):
                if first:
                    c.zero_()
                if blockops.is_sparse(bb) and \
                        not _route_dense(bb, out.device):
                    # sparse x sparse: CSR x CSR kernel, no densified-B
                    # FLOPs or bytes (VERDICT r1 missing #2)
                    if blockops.spgemm_acc(c, ab, bb):
                        continue
                blockops.spmm_acc(c, ab, blockops.densify(bb))
                continue
            bbd = blockops.densify(bb)
            bt = _bt_cached(b, k, bj, bbd)
            if mx8 and _mx8_usable(ab, mxq) and _mx8_usable(bt, mxq) \
                    and ab.shape[0] % 256 == 0 and bt.shape[0] % 256 == 0:
                a8, asc = _mx8_cached(a, (bi, k), ab, mxq)
                bt8, btsc = _mx8_cached(b, ("t", k, bj), bt, mxq)
                _mxq_gemm(mxq, c, a8, asc, bt8, btsc, beta0=first)
                continue
            if use_fused and last:
                fused_done = blockops.gemm_bt_final(c, outblk, ab, bt,
                                                    beta0=first)
                if not fused_done:
                    blockops.gemm_bt_acc(c, ab, bt, beta0=first)
            else:
                blockops.gemm_bt_acc(c, ab, bt, beta0=first)
        out.blocks[(bi, bj)] = outblk if fused_done else c.to(out.dtype)


def summa_matmul(a: BlockMatrix, b: BlockMatrix, out: BlockMatrix) -> None:
    """SUMMA over the (Pr x Pc) grid. a, b, out all GridScheme.

    For k-step k:
      * A panel = stack of blocks A(i, k) for this grid-row's i's, owned by
        grid column k % Pc -> broadcast over the row sub-group.
      * B panel = stack of blocks B(k, j) for this grid-col's j's, owned by
        grid row k % Pr -> broadcast over the col sub-group.
      * C(i, j) += A(i, k) @ B(k, j) locally (fp32 accumulators).
    Panels for step k+1 are prefetched async while step k computes.
    Fully-empty k-panels (block-sparse operands) are skipped.
    """
    rt = get_runtime()
    if not rt.is_distributed:
        local_matmul(a, b, out)
        return
    pr, pc = rt.grid
    p, q = rt.grid_row, rt.grid_col
    gk = a.grid_cols
    assert gk == b.grid_rows
    # Owner-side panel prep (pads/transposes/mx8 conversions/CSR packs)
    # runs on a second HIP stream so it overlaps the already-enqueued
    # GEMMs of the previous step; broadcasts wait on it via events.
    use_prep = (a.device.type == "cuda" and get_config().overlap_comm
                and get_config().prep_stream
                and not get_config().deterministic)
    prep = _get_prep_stream() if use_prep else None
    if prep is not None:
        _ev0 = torch.cuda.Event()
        _ev0.record()
        prep.wait_event(_ev0)  # prep ordered after operand production

    def _prep_ctx():
        import contextlib
        return (torch.cuda.stream(prep) if prep is not None
                else contextlib.nullcontext())

    def _prep_reg(*ts):
        if prep is not None:
            for t in ts:
                t.record_stream(prep)

    def _prep_done():
        if prep is not None:
            _e = torch.cuda.Event()
            _e.record(prep)
            torch.cuda.current_stream().wait_event(_e)

    pa, pb = _presence_map(a), _presence_map(b)
    # Global stored-nnz maps: drive the group-uniform dense-vs-CSR wire
    # decision for BOTH panel directions (element-sparse blocks broadcast
    # their CSR arrays - nnz-proportional xGMI bytes - instead of a
    # densified bs^2 payload). Must be called by every rank (collective,
    # cached).
    nma = a.nnz_map()
    nmb = b.nnz_map()
    sp_thr = get_config().sparse_threshold
    bs = a.block_size
    # MX-fp8/fp4 SUMMA: owners convert panels to quantized payloads +
    # scales ONCE and the broadcasts move 1/2 (fp8) or 1/4 (fp4) of the
    # bf16 bytes over xGMI (+ 1/32 for scales).
    _prec = get_config().gemm_precision
    mxq = (_prec if _prec in _MXQ_PRECS else None) if (
        a.device.type == "cuda" and a.dtype == torch.bfloat16
        and b.dtype == torch.bfloat16 and bs % 256 == 0) else None
    mx8 = mxq is not None
    pbytes = _mxq_pbytes(mxq, bs)  # payload bytes per row

    my_is = [i for i in range(out.grid_rows) if i % pr == p]
    my_js = [j for j in range(out.grid_cols) if j % pc == q]

    # fp32 accumulators for owned C blocks
    acc: Dict[Tuple[int, int], torch.Tensor] = {}

    # Broadcast participation must be UNIFORM across each sub-group
    # (collectives deadlock otherwise): the A-panel predicate depends only
    # on the grid ROW (my_is is shared by the whole row group), the
    # B-panel predicate only on the grid COLUMN. A rank therefore joins
    # the A broadcast at steps ka, the B broadcast at steps kb, and
    # computes at ka & kb.
    ka = {k for k in range(gk) if any(pa[i, k] for i in my_is)}
    kb = {k for k in range(gk) if any(pb[k, j] for j in my_js)}
    ks = sorted(ka | kb)
    if not ks:
        return

    def issue_bcast(k: int):
        """Stack panels for step k and issue async broadcasts.

        Only PRESENT blocks enter the panel (the presence maps are
        globally synced, so every rank in the sub-group derives the same
        stack layout) - a 1%-block-density operand broadcasts 1% of the
        bytes (config 4). A rank joins the A (B) broadcast only at steps
        in ka (kb) - group-uniform by construction.
        """
        a_rows, a_stack = [], None
        a_sc = None
        a_spinfo: Dict[int, tuple] = {}
        a_dn_index: Dict[int, int] = {}
        waits = []
        if k in ka:
            a_src = rt.rank_of(p, k % pc)
            a_rows = [i for i in my_is if pa[i, k]]
            a_dn_index = {i: n for n, i in enumerate(a_rows)}
            if mx8:
                a_stack = torch.empty(len(a_rows), bs, pbytes,
                                      dtype=torch.uint8, device=a.device)
                # packed-dword scales [bs/128][bs] int32 (mxconvert r2)
                a_sc = torch.empty(len(a_rows), bs // 128, bs,
                                   dtype=torch.int32, device=a.device)
                _prep_reg(a_stack, a_sc)
                if rt.rank == a_src:
                    with _prep_ctx():
                        for n, i in enumerate(a_rows):
                            p8, sc = _mx8_padded(a, ("s", i, k),
                                                 a.get_block_dense(i, k),
                                                 bs, mxq)
                            a_stack[n] = p8
                            a_sc[n] = sc
                    _prep_done()
                # NCCL executes ops of one communicator in enqueue order,
                # so waiting the (later) scale broadcast also fences the
                # payload broadcast.
                waits.append(dist.broadcast(a_stack, src=a_src,
                                            group=rt.row_group(p),
                                            async_op=True))
                waits.append(dist.broadcast(a_sc, src=a_src,
                                            group=rt.row_group(p),
                                            async_op=True))
            else:
                # dense-vs-CSR wire split, uniform across the row group
                # (derived from the globally-synced nnz map + geometry)
                sp_rows = [i for i in a_rows
                           if int(nma[i, k]) < sp_thr
                           * a.block_shape(i, k)[0] * a.block_shape(i, k)[1]]
                spset = set(sp_rows)
                dn_rows = [i for i in a_rows if i not in spset]
                a_dn_index = {i: n for n, i in enumerate(dn_rows)}
                # empty, not zeros: the broadcast overwrites it on
                # receivers; the source zero-pads only ragged edge blocks
                a_stack = torch.empty(len(dn_rows), bs, bs, dtype=a.dtype,
                                      device=a.device)
                _prep_reg(a_stack)
                if rt.rank == a_src:
                    with _prep_ctx():
                        for n, i in enumerate(dn_rows):
                            t = blockops.densify(a.get_block_dense(i, k))
                            h, w = t.shape
                            if (h, w) != (bs, bs):
                                a_stack[n].zero_()
                            a_stack[n, :h, :w] = t
                    _prep_done()
                if len(dn_rows):
                    waits.append(dist.broadcast(a_stack, src=a_src,
                                                group=rt.row_group(p),
                                                async_op=True))
                    add_comm("summa_a_bcast", a_stack.numel()
                             * a_stack.element_size())
                if sp_rows:
                    # concatenated CSR arrays, exact sizes known to every
                    # rank: per block [crow (h+1) | cols (nnz)] int32,
                    # values in the operand dtype.
                    shapes = {i: a.block_shape(i, k) for i in sp_rows}
                    nnzs = {i: int(nma[i, k]) for i in sp_rows}
                    idx_len = sum(shapes[i][0] + 1 + nnzs[i]
                                  for i in sp_rows)
                    val_len = sum(nnzs[i] for i in sp_rows)
                    idx_buf = torch.empty(idx_len, dtype=torch.int32,
                                          device=a.device)
                    val_buf = torch.empty(val_len, dtype=a.dtype,
                                          device=a.device)
                    _prep_reg(idx_buf, val_buf)
                    if rt.rank == a_src:
                        with _prep_ctx():
                            io = vo = 0
                            for i in sp_rows:
                                t = a.blocks[(i, k)]
                                if not blockops.is_sparse(t):
                                    t = t.to_sparse_csr()
                                h = shapes[i][0]
                                nz = nnzs[i]
                                idx_buf[io:io + h + 1] = \
                                    t.crow_indices().to(torch.int32)
                                idx_buf[io + h + 1:io + h + 1 + nz] = \
                                    t.col_indices().to(torch.int32)
                                val_buf[vo:vo + nz] = t.values()
                                io += h + 1 + nz
                                vo += nz
                        _prep_done()
                    waits.append(dist.broadcast(idx_buf, src=a_src,
                                                group=rt.row_group(p),
                                                async_op=True))
                    waits.append(dist.broadcast(val_buf, src=a_src,
                                                group=rt.row_group(p),
                                                async_op=True))
                    add_comm("summa_a_bcast_csr",
                             idx_buf.numel() * 4
                             + val_buf.numel() * val_buf.element_size())
                    # per-block views into the (in-flight) buffers; valid
                    # to read after the waits drain
                    io = vo = 0
                    for i in sp_rows:
                        h, w = shapes[i]
                        nz = nnzs[i]
                        a_spinfo[i] = (idx_buf[io:io + h + 1],
                                       idx_buf[io + h + 1:io + h + 1 + nz],
                                       val_buf[vo:vo + nz], h, w)
                        io += h + 1 + nz
                        vo += nz
        b_cols, b_stack = [], None
        b_sc = None
        b_spinfo: Dict[int, tuple] = {}
        b_dn_index: Dict[int, int] = {}
        if k in kb:
            # B panel is broadcast PRE-TRANSPOSED: the owner transposes
            # each block once (the B^T GEMM operand), so the other pc-1
            # ranks in the column group skip the transpose entirely and
            # receivers feed panels straight into the MFMA kernel.
            # Element-sparse B blocks go over the wire as CSR arrays
            # (UNtransposed; nnz-proportional bytes - the B-side mirror
            # of the A CSR wire, VERDICT r1 missing #3); receivers
            # reconstruct the dense transposed panel only when a dense-A
            # GEMM actually needs it.
            b_src = rt.rank_of(k % pr, q)
            b_cols = [j for j in my_js if pb[k, j]]
            b_dn_index = {j: n for n, j in enumerate(b_cols)}
            if mx8:
                b_stack = torch.empty(len(b_cols), bs, pbytes,
                                      dtype=torch.uint8, device=b.device)
                b_sc = torch.empty(len(b_cols), bs // 128, bs,
                                   dtype=torch.int32, device=b.device)
                _prep_reg(b_stack, b_sc)
                if rt.rank == b_src:
                    with _prep_ctx():
                        for n, j in enumerate(b_cols):
                            t = _pad_t(blockops.densify(
                                b.get_block_dense(k, j)), bs)
                            p8, sc = _mx8_padded(b, ("st", k, j),
                                                 blockops.transpose(t),
                                                 bs, mxq)
                            b_stack[n] = p8
                            b_sc[n] = sc
                    _prep_done()
                waits.append(dist.broadcast(b_stack, src=b_src,
                                            group=rt.col_group(q),
                                            async_op=True))
                waits.append(dist.broadcast(b_sc, src=b_src,
                                            group=rt.col_group(q),
                                            async_op=True))
            else:
                sp_cols = [j for j in b_cols
                           if int(nmb[k, j]) < sp_thr
                           * b.block_shape(k, j)[0] * b.block_shape(k, j)[1]]
                spset = set(sp_cols)
                dn_cols = [j for j in b_cols if j not in spset]
                b_dn_index = {j: n for n, j in enumerate(dn_cols)}
                b_stack = torch.empty(len(dn_cols), bs, bs, dtype=b.dtype,
                                      device=b.device)
                _prep_reg(b_stack)
                if rt.rank == b_src:
                    with _prep_ctx():
                        for n, j in enumerate(dn_cols):
                            t = blockops.densify(b.get_block_dense(k, j))
                            h, w = t.shape
                            if (h, w) != (bs, bs):
                                pad = torch.zeros(bs, bs, dtype=t.dtype,
                                                  device=t.device)
                                pad[:h, :w] = t
                                t = pad
                            if t.is_cuda:
                                b_stack[n] = blockops.transpose(t)
                            else:
                                b_stack[n] = t.t()
                    _prep_done()
                if len(dn_cols):
                    waits.append(dist.broadcast(b_stack, src=b_src,
                                                group=rt.col_group(q),
                                                async_op=True))
                    add_comm("summa_b_bcast", b_stack.numel()
                             * b_stack.element_size())
                if sp_cols:
                    shapes = {j: b.block_shape(k, j) for j in sp_cols}
                    nnzs = {j: int(nmb[k, j]) for j in sp_cols}
                    idx_len = sum(shapes[j][0] + 1 + nnzs[j]
                                  for j in sp_cols)
                    val_len = sum(nnzs[j] for j in sp_cols)
                    idx_buf = torch.empty(idx_len, dtype=torch.int32,
                                          device=b.device)
                    val_buf = torch.empty(val_len, dtype=b.dtype,
                                          device=b.device)
                    _prep_reg(idx_buf, val_buf)
                    if rt.rank == b_src:
                        with _prep_ctx():
                            io = vo = 0
                            for j in sp_cols:
                                t = b.blocks[(k, j)]
                                if not blockops.is_sparse(t):
                                    t = t.to_sparse_csr()
                                h = shapes[j][0]
                                nz = nnzs[j]
                                idx_buf[io:io + h + 1] = \
                                    t.crow_indices().to(torch.int32)
                                idx_buf[io + h + 1:io + h + 1 + nz] = \
                                    t.col_indices().to(torch.int32)
                                val_buf[vo:vo + nz] = t.values()
                                io += h + 1 + nz
                                vo += nz
                        _prep_done()
                    waits.append(dist.broadcast(idx_buf, src=b_src,
                                                group=rt.col_group(q),
                                                async_op=True))
                    waits.append(dist.broadcast(val_buf, src=b_src,
                                                group=rt.col_group(q),
                                                async_op=True))
                    add_comm("summa_b_bcast_csr",
                             idx_buf.numel() * 4
                             + val_buf.numel() * val_buf.element_size())
                    io = vo = 0
                    for j in sp_cols:
                        h, w = shapes[j]
                        nz = nnzs[j]
                        b_spinfo[j] = (idx_buf[io:io + h + 1],
                                       idx_buf[io + h + 1:io + h + 1 + nz],
                                       val_buf[vo:vo + nz], h, w)
                        io += h + 1 + nz
                        vo += nz
            if mx8 and b_stack is not None:
                add_comm("summa_b_bcast", b_stack.numel()
                         * b_stack.element_size())
        return {"a_rows": a_rows, "a_stack": a_stack, "a_sc": a_sc,
                "a_spinfo": a_spinfo, "a_dn": a_dn_index,
                "b_cols": b_cols, "b_stack": b_stack, "b_sc": b_sc,
                "b_spinfo": b_spinfo, "b_dn": b_dn_index,
                "waits": waits}

    # last contributing k per owned (i, j) - enables the fused bf16-out
    # epilogue on that call (no fp32 store + no conversion pass).
    kab = sorted(ka & kb)
    last_k: Dict[Tuple[int, int], int] = {}
    for k in kab:
        for i in my_is:
            if not pa[i, k]:
                continue
            for j in my_js:
                if pb[k, j]:
                    last_k[(i, j)] = k
    fused_ok = (out.device.type == "cuda" and out.dtype == torch.bfloat16
                and bs % 128 == 0)
    fused: Dict[Tuple[int, int], torch.Tensor] = {}

    # overlap_comm=True (default): panel broadcasts for the next
    # `summa_prefetch` k-steps are kept in flight while the current step
    # computes (RCCL's stream runs them under the compute; depth 1 =
    # classic double buffering, deeper costs depth x panel bytes of
    # memory). overlap_comm=False (deterministic/debug mode): each step's
    # broadcasts are issued only after the previous step's compute is
    # enqueued (serialized). Every rank issues in the same global order
    # (row-group then col-group, by ascending k), which keeps the
    # cross-communicator enqueue order NCCL-consistent at any depth.
    from collections import deque
    cfg_ = get_config()
    depth = max(1, cfg_.summa_prefetch) if cfg_.overlap_comm else 0
    pending: deque = deque()
    issued = 0

    def _fill(target: int) -> None:
        nonlocal issued
        while issued < len(ks) and len(pending) < target:
            pending.append((ks[issued], issue_bcast(ks[issued])))
            issued += 1

    _fill(1 + depth)
    for _step in range(len(ks)):
        k, cur = pending.popleft()
        if depth > 0:
            _fill(depth)  # keep `depth` panels in flight beyond cur
        a_rows, a_stack = cur["a_rows"], cur["a_stack"]
        a_sc, a_spinfo, a_dn = cur["a_sc"], cur["a_spinfo"], cur["a_dn"]
        b_cols, b_stack, b_sc = cur["b_cols"], cur["b_stack"], cur["b_sc"]
        b_spinfo, b_dn = cur["b_spinfo"], cur["b_dn"]
        for w_ in cur["waits"]:
            w_.wait()
        if not (k in ka and k in kb):
            if depth == 0:
                _fill(1)
            continue
        a_dense: Dict[int, torch.Tensor] = {}  # dense-routed CSR-A blocks
        for j in b_cols:
            bsp = b_spinfo.get(j)
            b_csr = None
            if bsp is None:
                bt = b_stack[b_dn[j]]  # transposed by the panel owner
            else:
                # CSR-wire B block: keep CSR; materialize the dense
                # transposed panel lazily, only if a dense-A GEMM below
                # actually needs it.
                crowb, colsb, valsb, hb, wb = bsp
                b_csr = torch.sparse_csr_tensor(
                    crowb, colsb, valsb, size=(hb, wb))
                bt = None
            b_unt = None  # lazily un-transposed B for CSR-A x dense-B

            def _need_bt():
                nonlocal bt, b_unt
                if bt is None:
                    d = _pad_t(blockops.densify(b_csr), bs)
                    b_unt = d
                    bt = (blockops.transpose(d) if d.is_cuda else
                          d.t().contiguous())
                return bt

            for i in a_rows:
                c = acc.get((i, j))
                first = c is None
                if first:
                    c = torch.empty(bs, bs, dtype=torch.float32,
                                    device=out.device)
                    acc[(i, j)] = c
                sp = a_spinfo.get(i)
                if sp is not None:
                    crow, cols, vals, h, w = sp
                    a_csr = torch.sparse_csr_tensor(
                        crow, cols, vals, size=(h, w))
                    if (out.device.type == "cuda" and len(vals)
                            >= get_config().spmm_densify_crossover * h * w):
                        # dense-route: CSR wire (nnz bytes) but MFMA
                        # compute (faster above the measured crossover)
                        ad = a_dense.get(i)
                        if ad is None:
                            ad = _pad_t(blockops.densify(a_csr), bs)
                            a_dense[i] = ad
                        bt_j = _need_bt() if bt is None else bt
                        blockops.gemm_bt_acc(c, ad, bt_j, beta0=first)
                        continue
                    if first:
                        c.zero_()
                    if b_csr is not None:
                        # both wires sparse: CSR x CSR block product
                        if blockops.spgemm_acc(c[:h], a_csr, b_csr):
                            continue
                    if b_unt is None:
                        _need_bt()
                        if b_unt is None:  # dense-wire B: transpose back
                            b_unt = (blockops.transpose(bt) if bt.is_cuda
                                     else bt.t())
                    blockops.spmm_acc(c[:h], a_csr, b_unt[:w])
                    continue
                n_i = a_dn[i]
                if mx8:
                    _mxq_gemm(mxq, c, a_stack[n_i], a_sc[n_i],
                              bt, b_sc[b_dn[j]], beta0=first)
                    continue
                bt_j = _need_bt() if bt is None else bt
                if fused_ok and last_k.get((i, j)) == k:
                    ob = torch.empty(bs, bs, dtype=out.dtype,
                                     device=out.device)
                    if blockops.gemm_bt_final(c, ob, a_stack[n_i], bt_j,
                                              beta0=first):
                        fused[(i, j)] = ob
                        continue
                blockops.gemm_bt_acc(c, a_stack[n_i], bt_j, beta0=first)
        if depth == 0:
            _fill(1)  # serialized mode: issue only after compute enqueued

    for (i, j), c in acc.items():
        h, w = out.block_shape(i, j)
        ob = fused.get((i, j))
        if ob is not None:
            out.blocks[(i, j)] = (ob if (h, w) == (bs, bs)
                                  else ob[:h, :w].contiguous())
        else:
            # contiguous: on CPU .to() with an unchanged dtype is a no-op
            # and a ragged crop stays a strided view, which a later
            # collective (to_dense gather, repartition) must not see
            out.blocks[(i, j)] = c[:h, :w].to(out.dtype).contiguous()


def bcast_matmul(a: BlockMatrix, b: BlockMatrix, out: BlockMatrix,
                 replicate: str) -> None:
    """Broadcast-MM: the small operand is replicated (BroadcastScheme),
    the large one row-(A)/col-(B)-partitioned; each C block is then
    computable locally. `replicate` in {'a','b'} names the small side.
    The executor has already repartitioned both operands."""
    local_matmul(a, b, out)
