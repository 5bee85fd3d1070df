import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a GPU (run on an MI355X box)")
    config.addinivalue_line(
        "markers", "slow: long-running CPU test (world-8 soaks etc.)")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU visible")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def free_port() -> int:
    """An OS-assigned free TCP port (avoids fixed-port collisions with
    lingering sockets from earlier test processes)."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]
