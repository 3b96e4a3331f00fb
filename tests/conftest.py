import os
import socket
from pathlib import Path

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run via gpurun)")
    config.addinivalue_line("markers", "slow: multi-process / long-running CPU test")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


def find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.fixture
def free_port():
    return find_free_port()


@pytest.fixture
def tiny_pbin(tmp_path) -> Path:
    """A small .pbin file with known contents."""
    from modalities_amd.dataloader.packed_data import write_pbin

    docs = [np.arange(i * 10, i * 10 + 20 + i, dtype=np.uint16) for i in range(8)]
    path = tmp_path / "tiny.pbin"
    write_pbin(path, docs, token_size_in_bytes=2)
    return path
