import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU on this box")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(autouse=True)
def reset_singletons():
    """Reset framework singletons between tests (the reference's
    AccelerateTestCase pattern, test_utils/testing.py:667-679)."""
    from accelerate_amd.state import AcceleratorState, GradientState, PartialState

    AcceleratorState._reset_state()
    PartialState._reset_state()
    GradientState._reset_state()
    yield
    AcceleratorState._reset_state()
    PartialState._reset_state()
    GradientState._reset_state()
