import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X GPU (run via gpurun)")
    config.addinivalue_line(
        "markers", "slow: multi-process integration test (>30s)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def free_port() -> int:
    """OS-assigned free TCP port for torch.distributed rendezvous.
    Fixed port numbers collide with TIME_WAIT sockets or straggler
    workers from a previous suite run and fail the store bind — seen
    once as a transient test_ddp_gloo failure."""
    import socket
    with socket.socket() as s:
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]
