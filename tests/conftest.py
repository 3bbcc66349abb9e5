import os

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X (run on a GPU box)")


def pytest_collection_modifyitems(config, items):
    """Auto-skip gpu tests when no device is present, so `pytest tests` works
    everywhere; the driver still selects with -m gpu / -m 'not gpu'."""
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:  # noqa: BLE001
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(autouse=True)
def _isolate_ipc_dir(tmp_path, monkeypatch):
    """Each test gets its own IPC socket dir and job name."""
    monkeypatch.setenv("DLROVER_IPC_SOCKET_DIR", str(tmp_path / "ipc"))
    yield
