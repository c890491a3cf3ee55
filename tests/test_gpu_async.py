"""GPU coverage for the TRUE-ASYNC engine's host-staged path: two ranks
SHARE one MI355X (both map cuda:0; the engine's payloads are host-staged
gloo p2p, so no cross-process device IPC is involved). This exercises
the shard-on-HBM <-> CPU staging conversions (async_ps.py) that
single-process GPU tests and CPU-only dist tests both miss. The real
multi-GPU RCCL plane is validated by the driver's 8-GPU scale run."""

import pytest
import torch

from conftest import run_dist

pytestmark = pytest.mark.gpu


def _async_gpu_shards(rank, world):
    import multiverso_amd as mv
    mv.init()   # async mode; device = cuda:0 on BOTH ranks
    zoo = mv.Zoo.get()
    assert zoo.device.type == "cuda"
    assert zoo.async_engine is not None

    # whole-table: GPU shard, host-staged slices, updater kernel on HBM
    t = mv.MatrixTable(1000, 32, updater_type="sgd")
    assert t.shard.is_cuda
    delta = torch.full((1000, 32), 1.0)
    t.add(delta)                      # ack => applied (sgd: w -= delta)
    got = t.get()
    assert got.is_cuda
    assert float(got.max()) <= -1.0 + 1e-6   # at least my own add
    mv.barrier()
    got = t.get()
    assert torch.equal(got.cpu(), torch.full((1000, 32), -float(world)))

    # keyed ops: gather/scatter kernels on the owned HBM rows
    t2 = mv.MatrixTable(64, 8)
    t2.add_rows([rank, 63 - rank], torch.ones(2, 8))
    mv.barrier()
    rows = t2.get_rows([0, 1, 62, 63])
    assert rows.is_cuda
    expect = torch.zeros(4, 8)
    expect[0] = expect[3] = 1.0
    expect[1] = expect[2] = 1.0      # 63-0=63, 63-1=62
    assert torch.equal(rows.cpu(), expect), (rank, rows)

    # sparse stale service with HBM shards (OP_GET_STALE)
    sp = mv.SparseMatrixTable(32, 4)
    cache = torch.zeros(32, 4)
    assert sp.get_into(cache) == 32
    assert sp.get_into(cache) == 0
    mv.barrier()   # fence: a fast peer's add must not precede the ==0
    sp.add_rows([rank, 31 - rank], torch.ones(2, 4))
    mv.barrier()
    n = sp.get_into(cache)
    assert n == 4, (rank, n)     # both ranks' adds stale for me
    assert torch.equal(cache[0], torch.ones(4))

    # unequal op counts with GPU shards (the async signature move)
    t3 = mv.ArrayTable(128)
    for _ in range(2 if rank == 0 else 5):
        t3.add(torch.ones(128))
        t3.get()
    mv.barrier()
    got = t3.get()
    assert torch.equal(got.cpu(), torch.full((128,), 7.0))
    mv.shutdown()


def test_async_engine_gpu_shards():
    run_dist(_async_gpu_shards, 2, gpu_share=True)


def _role_split_gpu(rank, world):
    """Dedicated-server topology with the shard in HBM: rank 1 hosts,
    rank 0 trains."""
    import multiverso_amd as mv
    mv.init([f"-ps_role={'server' if rank == 1 else 'worker'}"])
    t = mv.ArrayTable(256, updater_type="adagrad")
    if rank == 1:
        assert t.shard.is_cuda and t.shard.numel() == 256
    else:
        assert t.shard.numel() == 0
        from multiverso_amd import AddOption
        opt = AddOption(learning_rate=0.1, rho=0.1)
        t.add(torch.ones(256), option=opt)
        got = t.get()
        assert float(got.max()) < 0  # adagrad stepped negative
    mv.barrier()
    mv.shutdown()


def test_ps_role_split_gpu():
    run_dist(_role_split_gpu, 2, gpu_share=True)
