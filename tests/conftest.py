import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no GPU is present."""
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def fixed_clock():
    """Deterministic millisecond clock starting at a fixed epoch."""
    state = {"t": 1_700_000_000_000}

    def clock():
        state["t"] += 1000
        return state["t"]

    return clock


@pytest.fixture
def seq_uuid():
    """Deterministic uuid generator."""
    state = {"n": 0}

    def gen():
        state["n"] += 1
        return f"00000000-0000-4000-8000-{state['n']:012d}"

    return gen
