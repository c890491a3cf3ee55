import os
import socket
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)")
    # The CPU tier must behave identically on GPU-equipped hosts:
    # -m "not gpu" masks the devices before torch can initialize them.
    if "not gpu" in (config.getoption("-m") or ""):
        os.environ["HIP_VISIBLE_DEVICES"] = ""
        os.environ["CUDA_VISIBLE_DEVICES"] = ""


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _dist_entry(rank: int, fn, world_size: int, port: int, backend: str,
                gpu_share: bool = False):
    if backend == "gloo" and not gpu_share:
        # these are CPU-tier tests: mask any GPU so world_size ranks on a
        # 1-GPU host don't map LOCAL_RANK -> missing cuda devices
        os.environ["HIP_VISIBLE_DEVICES"] = ""
        os.environ["CUDA_VISIBLE_DEVICES"] = ""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    # gpu_share: every rank maps the SAME single GPU (cuda:0) — used to
    # exercise the host-staged async engine with CUDA shards on 1-GPU
    # boxes (the collective/RCCL plane needs real multi-GPU and is the
    # driver's job)
    os.environ["LOCAL_RANK"] = "0" if gpu_share else str(rank)
    os.environ["MV_BACKEND"] = backend
    fn(rank, world_size)


def run_dist(fn, world_size: int = 2, backend: str = "gloo",
             timeout: int = 120, gpu_share: bool = False):
    """Run ``fn(rank, world_size)`` in world_size fresh processes over gloo.

    This is the rebuild's analog of the reference's `mpirun -np N` CLI
    tests (SURVEY.md §4 tier 2).
    """
    import torch.multiprocessing as mp
    port = free_port()
    mp.start_processes(_dist_entry,
                       args=(fn, world_size, port, backend, gpu_share),
                       nprocs=world_size, join=True, start_method="spawn")
