import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)")
    config.addinivalue_line(
        "markers", "integration: slower multi-process integration test")


def pytest_collection_modifyitems(config, items):
    """Skip gpu tests automatically when no GPU is present."""
    import torch
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def tmp_resource_spec(tmp_path):
    """A 1-node CPU resource spec yaml (reference tests/test_autodist.py:5-21)."""
    p = tmp_path / "resource_spec.yml"
    p.write_text(
        "nodes:\n  - address: 127.0.0.1\n    cpus: [0]\n    chief: true\n")
    return str(p)


@pytest.fixture
def tmp_gpu_resource_spec(tmp_path):
    """A synthetic 8-GPU single-node spec (the MI355X target topology)."""
    p = tmp_path / "resource_spec_gpu.yml"
    p.write_text(
        "nodes:\n  - address: 127.0.0.1\n    gpus: [0,1,2,3,4,5,6,7]\n"
        "    cpus: [0]\n    chief: true\n")
    return str(p)
