"""On-disk block-matrix format: save/load round trip, COO import."""

import os

import torch

from matrel_amd.io.format import load_coo_text, load_matrix, save_matrix


def test_save_load_roundtrip(session, tmp_path):
    A = session.randn(100, 70, seed=1)
    path = str(tmp_path / "mat")
    A.save(path)
    assert os.path.exists(os.path.join(path, "manifest.json"))
    B = session.load(path)
    assert torch.allclose(A.to_tensor(), B.to_tensor(), atol=0)
    # manifest sanity
    import json
    man = json.load(open(os.path.join(path, "manifest.json")))
    assert man["nrows"] == 100 and man["ncols"] == 70
    assert man["block_size"] == 48
    assert len(man["blocks"]) == 6  # ceil(100/48) x ceil(70/48) = 3 x 2


def test_save_load_block_sparse(session, tmp_path):
    A = session.block_sparse_random(200, 200, 0.3, seed=3)
    path = str(tmp_path / "sparse")
    A.save(path)
    B = session.load(path)
    assert torch.allclose(A.to_tensor(), B.to_tensor(), atol=0)
    # absent blocks stay absent
    assert len(B.blocks.blocks) == len(A.blocks.blocks)


def test_checkpoint_resume_compute(session, tmp_path):
    """save/load doubles as checkpoint/resume."""
    A = session.randn(90, 90, seed=4)
    B = session.randn(90, 90, seed=5)
    C = (A @ B).compute()
    path = str(tmp_path / "ckpt")
    C.save(path)
    C2 = session.load(path)
    D = (C2 + C2).compute()
    assert torch.allclose(D.to_tensor(), 2 * C.to_tensor(), atol=1e-5)


def test_coo_text_import(session, tmp_path):
    p = tmp_path / "coo.txt"
    p.write_text("# i j v\n0 0 1.5\n10 20 -2.0\n99 49 3.25\n")
    A = session.load_coo(str(p), 100, 50)
    t = A.to_tensor()
    assert t[0, 0] == 1.5
    assert t[10, 20] == -2.0
    assert t[99, 49] == 3.25
    assert t.count_nonzero() == 3


def test_ones_eye_factories(session):
    import torch
    A = session.ones(70, 50)
    assert torch.equal(A.to_tensor(), torch.ones(70, 50))
    I = session.eye(100)
    assert torch.equal(I.to_tensor(), torch.eye(100))
    # identity is block-sparse: off-diagonal blocks absent
    assert len(I.blocks.blocks) == I.blocks.grid_rows
    B = session.randn(100, 60, seed=1)
    out = (I @ B).compute()
    assert torch.allclose(out.to_tensor(), B.to_tensor(), atol=1e-5)


def test_coo_roundtrip(session, tmp_path):
    import torch
    t = torch.zeros(60, 40)
    t[3, 4] = 1.5
    t[59, 39] = -2.25
    A = session.from_dense(t)
    p = str(tmp_path / "out.coo")
    A.save_coo(p)
    B = session.load_coo(p, 60, 40)
    assert torch.equal(B.to_tensor(), t)


def test_save_load_elem_sparse_csr(session, tmp_path):
    """CSR-stored blocks go to disk in CSR layout (nnz-proportional
    bytes) and come back CSR-stored with exact values."""
    import os
    A = session.sparse_random(300, 260, 0.02, seed=8)
    bm = A.blocks
    n_csr = sum(1 for _, t in bm.local_items()
                if t.layout == torch.sparse_csr)
    assert n_csr > 0
    path = str(tmp_path / "csr_mat")
    A.save(path)
    # on-disk bytes must be far below the dense footprint
    disk = sum(os.path.getsize(os.path.join(path, "blocks", f))
               for f in os.listdir(os.path.join(path, "blocks")))
    dense_bytes = 300 * 260 * 4
    assert disk < 0.4 * dense_bytes, (disk, dense_bytes)
    B = session.load(path)
    n_csr2 = sum(1 for _, t in B.blocks.local_items()
                 if t.layout == torch.sparse_csr)
    assert n_csr2 == n_csr
    assert torch.allclose(A.to_tensor(), B.to_tensor(), atol=0)
