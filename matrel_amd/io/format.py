"""On-disk block-matrix format + COO text import.

Reference equivalent: MatRel's Kryo block serializer + HDFS text/sequence
file save/load (SURVEY.md §2.1 "Serialization / on-disk format", §5
"Checkpoint / resume": the on-disk block-matrix format is a first-class
deliverable).

Layout of a saved matrix at <path>/:
    manifest.json            - dims, block size, dtype, scheme, and the
                               block table [{bi, bj, file, shape, nnz}]
    blocks/b_<bi>_<bj>.bin   - raw row-major little-endian tensor bytes
                               (mmap-able; dtype from the manifest)

save/load double as checkpoint/resume. In distributed runs every rank
writes its own blocks (shared filesystem), rank 0 merges the manifest;
load() reads only the blocks the calling rank owns under the target
scheme.
"""

from __future__ import annotations

import json
import os
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

from ..blocks.matrix import BlockMatrix
from ..blocks.schemes import Scheme, scheme_from_name
from ..exec.runtime import get_runtime

FORMAT_VERSION = 1

_NP_DTYPE = {
    torch.bfloat16: np.uint16,  # raw bf16 bits
    torch.float32: np.float32,
    torch.float64: np.float64,
    torch.float16: np.float16,
}
_DTYPE_NAME = {
    torch.bfloat16: "bf16", torch.float32: "fp32",
    torch.float64: "fp64", torch.float16: "fp16",
}
_NAME_DTYPE = {v: k for k, v in _DTYPE_NAME.items()}


def _block_file(bi: int, bj: int) -> str:
    return f"blocks/b_{bi}_{bj}.bin"


def save_matrix(m: BlockMatrix, path: str) -> None:
    rt = get_runtime()
    os.makedirs(os.path.join(path, "blocks"), exist_ok=True)
    entries = []
    for (bi, bj), t in m.local_items():
        if m.owner(bi, bj) == -1 and rt.rank != 0:
            continue  # replicated blocks written once
        fn = _block_file(bi, bj)
        if t.layout == torch.sparse_csr:
            # CSR on disk: [crow int32 (h+1) | col int32 (nnz) | values]
            # in one file - nnz-proportional bytes (the reference's
            # SparseMatrix serialization analog); offsets derive from
            # shape+nnz, so the manifest needs no extra fields.
            h, w = t.shape
            nnz = int(t._nnz())
            with open(os.path.join(path, fn), "wb") as f:
                t.crow_indices().cpu().to(torch.int32).numpy().tofile(f)
                t.col_indices().cpu().to(torch.int32).numpy().tofile(f)
                vals = t.values().detach().cpu().contiguous()
                raw = (vals.view(torch.uint16)
                       if vals.dtype == torch.bfloat16 else vals)
                raw.numpy().tofile(f)
            entries.append({
                "bi": bi, "bj": bj, "file": fn, "layout": "csr",
                "shape": [int(h), int(w)], "nnz": nnz,
            })
            continue
        cpu = t.detach().cpu().contiguous()
        raw = cpu.view(torch.uint16) if cpu.dtype == torch.bfloat16 else cpu
        arr = raw.numpy()
        arr.tofile(os.path.join(path, fn))
        entries.append({
            "bi": bi, "bj": bj, "file": fn,
            "shape": list(t.shape),
            "nnz": int(cpu.float().count_nonzero().item()),
        })
    if rt.is_distributed:
        gathered = [None] * rt.world_size
        dist.all_gather_object(gathered, entries)
        entries = [e for part in gathered for e in part]
        # replicated blocks may appear once (rank0) - no dedup needed
    if rt.rank == 0:
        manifest = {
            "format_version": FORMAT_VERSION,
            "nrows": m.nrows, "ncols": m.ncols,
            "block_size": m.block_size,
            "dtype": _DTYPE_NAME[m.dtype],
            "scheme": m.scheme.kind,
            "blocks": sorted(entries, key=lambda e: (e["bi"], e["bj"])),
        }
        with open(os.path.join(path, "manifest.json"), "w") as f:
            json.dump(manifest, f, indent=1)
    rt.barrier()


def load_matrix(path: str, scheme: Optional[Scheme] = None,
                device: Optional[torch.device] = None,
                mmap: bool = True) -> BlockMatrix:
    with open(os.path.join(path, "manifest.json")) as f:
        man = json.load(f)
    assert man["format_version"] == FORMAT_VERSION
    dtype = _NAME_DTYPE[man["dtype"]]
    sch = scheme or scheme_from_name(man["scheme"])
    out = BlockMatrix(man["nrows"], man["ncols"], man["block_size"],
                      sch, dtype, device)
    total_nnz = 0
    for e in man["blocks"]:
        total_nnz += e["nnz"]
        bi, bj = e["bi"], e["bj"]
        if not out.is_local(bi, bj):
            continue
        fp = os.path.join(path, e["file"])
        np_dt = _NP_DTYPE[dtype]
        if e.get("layout") == "csr":
            h, w = e["shape"]
            nnz = e["nnz"]
            raw = np.fromfile(fp, dtype=np.uint8)
            crow = torch.from_numpy(
                raw[:4 * (h + 1)].view(np.int32).copy())
            cols = torch.from_numpy(
                raw[4 * (h + 1):4 * (h + 1 + nnz)].view(np.int32).copy())
            vb = raw[4 * (h + 1 + nnz):].view(np_dt).copy()
            vals = torch.from_numpy(vb)
            if dtype == torch.bfloat16:
                vals = vals.view(torch.bfloat16)
            t = torch.sparse_csr_tensor(
                crow.to(out.device), cols.to(out.device),
                vals.to(device=out.device, dtype=dtype), size=(h, w))
            out.blocks[(bi, bj)] = t
            continue
        arr = np.memmap(fp, dtype=np_dt, mode="r") if mmap else \
            np.fromfile(fp, dtype=np_dt)
        t = torch.from_numpy(np.ascontiguousarray(arr)).reshape(e["shape"])
        if dtype == torch.bfloat16:
            t = t.view(torch.bfloat16)
        out.blocks[(bi, bj)] = t.to(device=out.device)
    out._density_hint = total_nnz / max(1, man["nrows"] * man["ncols"])
    return out


def load_coo_text(path: str, nrows: int, ncols: int,
                  **kw) -> BlockMatrix:
    """Import 'i j v' text triplets (the reference's COO reader,
    SURVEY.md §3.2). Indices are 0-based."""
    out = BlockMatrix(nrows, ncols, **kw)
    bs = out.block_size
    staged = {}
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line or line.startswith("#"):
                continue
            parts = line.replace(",", " ").split()
            i, j, v = int(parts[0]), int(parts[1]), float(parts[2])
            bi, bj = i // bs, j // bs
            if not out.is_local(bi, bj):
                continue
            key = (bi, bj)
            if key not in staged:
                staged[key] = torch.zeros(out.block_shape(bi, bj),
                                          dtype=torch.float32)
            staged[key][i - bi * bs, j - bj * bs] = v
    for key, t in staged.items():
        out.blocks[key] = t.to(device=out.device, dtype=out.dtype)
    return out


def save_coo_text(m: BlockMatrix, path: str) -> None:
    """Export nonzeros as 'i j v' text (the reference's COO text format).
    Rank 0 writes; intended for small/sparse matrices."""
    rt = get_runtime()
    full = m.to_dense()
    if rt.rank == 0:
        nz = full.nonzero(as_tuple=False).cpu()
        vals = full[full != 0].cpu()
        with open(path, "w") as f:
            f.write("# i j v\n")
            for (i, j), v in zip(nz.tolist(), vals.tolist()):
                f.write(f"{i} {j} {v}\n")
    rt.barrier()
