"""Multi-process (world_size=2, gloo) integration tests — the rebuild's
equivalent of the reference's `mpirun -np N` CLI tests (Test/, SURVEY.md §4
tier 2). Exact-value oracles scaled by world size follow the canonical
pattern of Test/test_array_table.cpp:31-42 and
binding/python tests (test_multiverso.py:33)."""

import numpy as np
import pytest
import torch

from conftest import run_dist


# ---- worker functions (must be top-level for spawn pickling) ----

def _array_sync_oracle(rank, world):
    import multiverso_amd as mv
    mv.init(sync=True)
    t = mv.ArrayTable(10)
    for it in range(3):
        t.add(torch.full((10,), 2.0)).wait()
        got = t.get()
        expect = torch.full((10,), 2.0 * (it + 1) * world)
        assert torch.equal(got, expect), (rank, it, got)
    mv.shutdown()


def _array_uneven(rank, world):
    import multiverso_amd as mv
    mv.init(sync=True)
    t = mv.ArrayTable(11)  # uneven: shards 5 + 6
    t.add(torch.arange(11, dtype=torch.float32))
    got = t.get()
    assert torch.equal(got, torch.arange(11, dtype=torch.float32) * world)
    mv.shutdown()


def _matrix_whole_and_rows(rank, world):
    import multiverso_amd as mv
    mv.init(sync=True)
    t = mv.MatrixTable(9, 4)  # uneven rows: 4 + 5
    delta = torch.ones(9, 4)
    t.add(delta)
    got = t.get()
    assert torch.equal(got, torch.full((9, 4), float(world)))
    # row ops: each rank adds to different rows
    rows = [rank, 8 - rank]
    t.add_rows(rows, torch.full((2, 4), 10.0))
    got = t.get_rows([0, 1, 8])
    expect = torch.full((3, 4), float(world))
    expect[0] += 10.0          # row 0 touched by rank 0
    expect[1] += 10.0          # row 1 touched by rank 1
    expect[2] += 10.0          # row 8 touched by rank 0 (8-0)
    if world == 1:
        expect = torch.full((3, 4), 1.0)
        expect[0] += 20.0
    assert torch.equal(got, expect), (rank, got)
    mv.shutdown()


def _matrix_same_rows(rank, world):
    import multiverso_amd as mv
    mv.init(sync=True)
    t = mv.MatrixTable(6, 2)
    # every rank adds the same rows -> sums
    t.add_rows([2, 5], torch.ones(2, 2))
    got = t.get_rows([2, 5])
    assert torch.equal(got, torch.full((2, 2), float(world)))
    mv.shutdown()


def _kv(rank, world):
    # default mode is TRUE ASYNC: my own adds are visible immediately
    # (FIFO per pair); everyone's adds only after a drain barrier
    import multiverso_amd as mv
    mv.init()
    t = mv.KVTable()
    t.add([1, 2, 3], [1.0, 2.0, 3.0])
    got = t.get([1])
    assert got[1] >= 1.0, got
    mv.barrier()
    got = t.get([1, 2, 3])
    assert got == {1: 1.0 * world, 2: 2.0 * world, 3: 3.0 * world}
    mv.shutdown()


def _aggregate(rank, world):
    import multiverso_amd as mv
    mv.init()
    x = torch.ones(5, dtype=torch.int32)
    mv.aggregate(x)
    assert torch.equal(x, torch.full((5,), world, dtype=torch.int32))
    mv.shutdown()


def _handler_init_protocol(rank, world):
    import multiverso_amd as mv
    mv.init(sync=True)
    h = mv.ArrayTableHandler(8, init_value=np.full(8, 3.0, dtype=np.float32))
    got = h.get()
    # master added init, others zeros -> value present exactly once
    assert torch.equal(got, torch.full((8,), 3.0)), (rank, got)
    mv.shutdown()


def _async_pipeline(rank, world):
    # default (sync=False) is the TRUE ASYNC mode at world>1: my get is
    # only guaranteed to see MY prior adds; a barrier (drain) makes
    # everyone's adds visible (reference MV_Barrier semantics)
    import multiverso_amd as mv
    mv.init()
    t = mv.ArrayTable(16)
    for _ in range(4):
        t.add(torch.ones(16), async_op=True)
    got = t.get()  # FIFO per server: sees at least my own 4 adds
    assert float(got.min()) >= 4.0, got
    mv.barrier()
    got = t.get()
    assert torch.equal(got, torch.full((16,), 4.0 * world))
    mv.shutdown()


# ---- tests ----

def test_array_sync_oracle():
    run_dist(_array_sync_oracle, 2)


def test_array_uneven_shards():
    run_dist(_array_uneven, 2)


def test_matrix_whole_and_rows():
    run_dist(_matrix_whole_and_rows, 2)


def test_matrix_same_rows():
    run_dist(_matrix_same_rows, 2)


def test_kv():
    run_dist(_kv, 2)


def test_aggregate():
    run_dist(_aggregate, 2)


def test_handler_init_protocol():
    run_dist(_handler_init_protocol, 2)


def test_async_pipeline():
    run_dist(_async_pipeline, 2)


def _net_bind_connect(rank, world, port):
    # explicit rendezvous: no MASTER_ADDR/RANK/WORLD_SIZE in env
    for k in ("MASTER_ADDR", "MASTER_PORT", "RANK", "WORLD_SIZE",
              "LOCAL_RANK"):
        import os
        os.environ.pop(k, None)
    import multiverso_amd as mv
    eps = [f"127.0.0.1:{port}", f"127.0.0.1:{port + 1}"]
    assert mv.net_bind(rank, eps[rank])
    assert mv.net_connect([0, 1], eps)
    mv.init(sync=True)
    assert mv.size() == world and mv.rank() == rank
    t = mv.ArrayTable(6)
    t.add(torch.full((6,), 3.0)).wait()
    assert torch.equal(t.get(), torch.full((6,), 3.0 * world))
    mv.shutdown()


def test_net_bind_connect():
    """MV_NetBind/MV_NetConnect deployment mode (multiverso.cpp:58-68):
    rendezvous from an explicit endpoint map, no launcher env."""
    import torch.multiprocessing as mp
    from conftest import free_port
    port = free_port()
    mp.start_processes(_net_bind_entry, args=(port,), nprocs=2,
                       join=True, start_method="spawn")


def _net_bind_entry(rank, port):
    _net_bind_connect(rank, 2, port)


# ---- world_size=4: the partition math and collective paths the driver
# exercises at N=4/8 GPUs (uneven shards: 11 = 2+2+2+5, 9 rows = 2+2+2+3)

def test_array_sync_oracle_ws4():
    run_dist(_array_sync_oracle, 4)


def test_array_uneven_shards_ws4():
    run_dist(_array_uneven, 4)


def test_array_uneven_shards_ws3():
    # odd world: 11 elements shard as 3+3+5 (remainder on the last server)
    run_dist(_array_uneven, 3)


def test_matrix_whole_and_rows_ws4():
    run_dist(_matrix_whole_and_rows, 4)


# ---- world_size=8: the exact rank count of the driver's 8-GPU node.
# One combined collective-path test (array oracle + keyed rows + sparse
# stale get + aggregate) keeps the 8-process spawn cost down.

def _all_paths_ws8(rank, world):
    import multiverso_amd as mv
    mv.init(sync=True)
    # array BSP oracle (test_array_table.cpp:31-42 pattern)
    a = mv.ArrayTable(19)            # uneven: 2x7 + 5 at ws8
    for it in range(2):
        a.add(torch.full((19,), 2.0)).wait()
        got = a.get()
        assert torch.equal(got, torch.full((19,), 2.0 * (it + 1) * world))
    # matrix whole + keyed (every rank different rows; some collide)
    m = mv.MatrixTable(21, 3)        # 2 rows x 7 + 7 on the last
    m.add(torch.ones(21, 3))
    m.add_rows([rank, 20 - rank, 10], torch.ones(3, 3))
    got = m.get_rows(list(range(21)))
    expect = torch.full((21, 3), float(world))
    for r in range(world):
        expect[r] += 1.0
        expect[20 - r] += 1.0
    expect[10] += world
    assert torch.equal(got, expect), (rank, got[:, 0])
    # sparse stale-filtered get
    s = mv.SparseMatrixTable(17, 2)
    cache = torch.zeros(17, 2)
    assert s.get_into(cache) == 17
    assert s.get_into(cache) == 0
    if rank == 3:
        s.add_rows([0, 16], torch.ones(2, 2))
    else:
        s.add_rows([], torch.zeros(0, 2))
    assert s.get_into(cache) == 2
    assert torch.equal(cache[16], torch.ones(2))
    # aggregate
    x = torch.ones(3)
    mv.aggregate(x)
    assert torch.equal(x, torch.full((3,), float(world)))
    mv.shutdown()


def test_all_collective_paths_ws8():
    run_dist(_all_paths_ws8, 8)


def _async_ws8(rank, world):
    import multiverso_amd as mv
    mv.init()
    t = mv.ArrayTable(16)
    for _ in range(rank + 1):        # every rank a DIFFERENT op count
        t.add(torch.ones(16))
        t.get()
    mv.barrier()
    got = t.get()
    total = sum(r + 1 for r in range(world))
    assert torch.equal(got, torch.full((16,), float(total))), (rank, got)
    mv.shutdown()


def test_async_ws8():
    run_dist(_async_ws8, 8)


def _kv_large(rank, world):
    """KV with a large key set (tensor exchange, not pickle): 20k keys
    per rank, half overlapping."""
    import multiverso_amd as mv
    mv.init(sync=True)
    t = mv.KVTable()
    base = 10_000 * rank
    keys = list(range(base, base + 20_000))
    t.add(keys, [1.0] * len(keys))
    got = t.get([5_000, 10_000 + 5_000 * (world - 1), 999_999])
    # overlap region [10000, 20000) got world contributions at ws2
    assert got[5_000] == 1.0
    if world == 2:
        assert got[15_000] == 2.0
    assert got[999_999] == 0.0
    mv.shutdown()


def test_kv_large_keyset():
    run_dist(_kv_large, 2)
