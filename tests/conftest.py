import os
import socket
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run on the GPU box)")


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU on this host")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _mp_entry(rank, world_size, port, fn, args):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import triton_dist_amd as td

    td.initialize_distributed()
    try:
        fn(rank, world_size, *args)
    finally:
        td.finalize_distributed()


def run_distributed(fn, world_size=2, args=(), timeout_s=120):
    """Run fn(rank, world_size, *args) in `world_size` spawned processes with
    a gloo (CPU) or RCCL (GPU) process group."""
    import torch.multiprocessing as mp

    port = free_port()
    mp.spawn(_mp_entry, args=(world_size, port, fn, args),
             nprocs=world_size, join=True)


@pytest.fixture
def dist_runner():
    return run_distributed
