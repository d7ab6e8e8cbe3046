import os
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
if str(REPO) not in sys.path:
    sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


# ---------------------------------------------------------------------------
# multi-process gloo harness (world_size>1 on CPU, 127.0.0.1 rendezvous)
# ---------------------------------------------------------------------------

def _dist_worker(rank, world_size, store_path, fn, args):
    import torch.distributed as dist
    store = dist.FileStore(store_path, world_size)
    dist.init_process_group("gloo", store=store, rank=rank,
                            world_size=world_size)
    try:
        fn(rank, world_size, *args)
    finally:
        dist.destroy_process_group()


def run_distributed(fn, world_size, *args, timeout=180):
    """Run ``fn(rank, world_size, *args)`` in world_size processes (gloo)."""
    import tempfile
    import torch.multiprocessing as mp
    with tempfile.TemporaryDirectory() as td:
        store_path = os.path.join(td, "store")
        mp.spawn(_dist_worker, args=(world_size, store_path, fn, args),
                 nprocs=world_size, join=True)


@pytest.fixture
def dist_runner():
    return run_distributed
