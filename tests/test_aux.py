"""Aux subsystems: checkpointed loop resume, hipGraph compile fallback,
tracing output, metrics report, debug env."""

import json
import os

import pytest
import torch


def test_checkpointed_loop_resume(session, tmp_path):
    from matrel_amd.exec.faults import CheckpointedLoop
    calls = []

    def step(i, state):
        calls.append(i)
        m = session.randn(64, 64, seed=i).blocks
        return {"w": m}

    loop = CheckpointedLoop(str(tmp_path / "ck"), every=2)
    loop.run(step, 4)
    assert calls == [0, 1, 2, 3]
    # resume: state restored at step 4 -> no further calls
    calls.clear()
    loop2 = CheckpointedLoop(str(tmp_path / "ck"), every=2)
    start, state = loop2.resume()
    assert start == 4 and "w" in state
    ref = session.randn(64, 64, seed=3).blocks.to_dense()
    assert torch.allclose(state["w"].to_dense(), ref, atol=0)


def test_compile_eager_fallback(session):
    A = session.randn(64, 64, seed=1)
    B = session.randn(64, 64, seed=2)
    fn = (A @ B + A).compile()  # CPU -> eager fallback
    out1 = fn()
    out2 = fn()
    ref = A.to_tensor() @ B.to_tensor() + A.to_tensor()
    assert torch.allclose(out1.to_tensor(), ref, atol=1e-3)
    assert torch.allclose(out2.to_tensor(), ref, atol=1e-3)


def test_tracing_chrome_output(session, tmp_path):
    from matrel_amd.exec import tracing
    tracing.clear()
    tracing.enable(True)
    A = session.randn(64, 64, seed=3)
    (A @ A.T).sum().compute()
    p = str(tmp_path / "trace.json")
    tracing.dump_trace(p)
    tracing.enable(False)
    ev = json.load(open(p))["traceEvents"]
    names = {e["name"] for e in ev}
    assert "MatMul" in names and "Agg" in names


@pytest.mark.gpu
def test_tracing_gpu_hipevent_spans(gpu_session, tmp_path):
    """hipEvent per-op GPU spans: tid-1 '[gpu]' events
    with nonzero duration on the same timeline as the host spans, no
    MATREL_TRACE_SYNC serialization."""
    from matrel_amd.exec import tracing
    tracing.clear()
    tracing.enable(True)
    s = gpu_session
    A = s.randn(1024, 1024, seed=4)
    (A @ A.T).sum().compute()
    p = str(tmp_path / "trace_gpu.json")
    tracing.dump_trace(p)
    tracing.enable(False)
    ev = json.load(open(p))["traceEvents"]
    gpu = [e for e in ev if e.get("tid") == 1 and e.get("ph") == "X"]
    host = [e for e in ev if e.get("tid") == 0 and e.get("ph") == "X"]
    assert any("MatMul" in e["name"] for e in gpu), [e["name"] for e in gpu]
    mm_gpu = next(e for e in gpu if "MatMul" in e["name"])
    assert mm_gpu["dur"] > 0
    # same timeline: GPU span lies within ~the host trace window
    t0 = min(e["ts"] for e in host) - 1e6
    t1 = max(e["ts"] + e["dur"] for e in host) + 1e6
    assert t0 <= mm_gpu["ts"] <= t1
    tracing.clear()


def test_metrics_report(session):
    session.reset_metrics()
    A = session.randn(64, 64, seed=4)
    (A @ A).compute()
    rep = session.metrics_report()
    assert rep["ops"]["MatMul"]["count"] >= 1
    assert rep["ops"]["MatMul"]["est_flops"] > 0


def test_debug_env(session):
    from matrel_amd.utils.debug import debug_env_report, serialize_kernels
    serialize_kernels(True)
    assert os.environ["AMD_SERIALIZE_KERNEL"] == "3"
    serialize_kernels(False)
    assert "AMD_SERIALIZE_KERNEL" not in os.environ
    assert isinstance(debug_env_report(), dict)


def test_config_from_env(monkeypatch):
    from matrel_amd.config import MatrelConfig
    monkeypatch.setenv("MATREL_BLOCK_SIZE", "2048")
    monkeypatch.setenv("MATREL_GEMM_PRECISION", "mx8")
    monkeypatch.setenv("MATREL_SPARSE_THRESHOLD", "0.1")
    monkeypatch.setenv("MATREL_OVERLAP_COMM", "0")
    monkeypatch.setenv("MATREL_DETERMINISTIC", "1")
    monkeypatch.setenv("MATREL_GRID", "4x2")
    cfg = MatrelConfig.from_env()
    assert cfg.block_size == 2048
    assert cfg.gemm_precision == "mx8"
    assert cfg.sparse_threshold == 0.1
    assert cfg.overlap_comm is False
    assert cfg.deterministic is True
    assert cfg.grid == (4, 2)
