import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X) and the built "
        "HIP extension; run with -m gpu on a GPU box")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture()
def session():
    """Single-process CPU session with small blocks so multi-block paths
    are exercised (BASELINE.json config 1 regime)."""
    import matrel_amd as ma
    from matrel_amd.exec.runtime import reset_runtime
    from matrel_amd.plan.logical import Leaf
    reset_runtime()
    cfg = ma.MatrelConfig(block_size=48)
    s = ma.MatrelSession(cfg)
    yield s
    reset_runtime()


@pytest.fixture()
def gpu_session():
    import matrel_amd as ma
    from matrel_amd.exec.runtime import reset_runtime
    reset_runtime()
    cfg = ma.MatrelConfig(block_size=512)
    s = ma.MatrelSession(cfg)
    yield s
    reset_runtime()
