import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run on the GPU box)")


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def tmp_datastore(tmp_path, monkeypatch):
    """Isolated datastore root for flow runs."""
    root = tmp_path / ".mfx"
    root.mkdir()
    monkeypatch.setenv("MFX_DATASTORE_SYSROOT_LOCAL", str(root))
    return str(root)
