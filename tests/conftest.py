import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)"
    )


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no GPU is present."""
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture()
def library_path(tmp_path, monkeypatch):
    d = tmp_path / "udp_lib"
    d.mkdir()
    monkeypatch.setenv("SATURN_LIBRARY_PATH", str(d))
    return str(d)


@pytest.fixture()
def save_dir(tmp_path):
    d = tmp_path / "saved_models"
    d.mkdir()
    return str(d)
