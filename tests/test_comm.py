"""Direct tests of the collective data-plane primitives (the rebuild's
equivalent of Test/test_net.cpp raw-transport tests): ShardSpec math,
padded uneven all-gather/reduce-scatter, keyed all-to-all routing, and
the model-average (-ma) mode."""

import pytest
import torch

import multiverso_amd as mv
from conftest import run_dist
from multiverso_amd.comm import ShardSpec


def test_shard_spec_even():
    s = ShardSpec(100, 4)
    assert s.counts == [25, 25, 25, 25] and s.even
    assert s.range_of(2) == (50, 25)
    assert s.owner_of(0) == 0 and s.owner_of(99) == 3


def test_shard_spec_uneven_remainder_to_last():
    # reference array_table.cpp:11-21: size/num_servers each, last gets
    # the remainder
    s = ShardSpec(10, 3)
    assert s.counts == [3, 3, 4] and not s.even
    assert s.offsets == [0, 3, 6]
    assert s.owner_of(9) == 2
    # owner clamps to last server (matrix_table.cpp:276)
    s2 = ShardSpec(7, 2)
    assert s2.owner_of(6) == 1


def _gather_scatter_uneven(rank, world):
    import torch
    import multiverso_amd as mv
    from multiverso_amd.comm import (ShardSpec, allgather_shards,
                                     reduce_scatter_delta)
    mv.init()
    spec = ShardSpec(7, world)  # uneven
    off, cnt = spec.range_of(rank)
    shard = torch.arange(off, off + cnt, dtype=torch.float32)
    out = torch.zeros(7)
    allgather_shards(out, shard, spec, 1)
    assert torch.equal(out, torch.arange(7, dtype=torch.float32)), out
    delta = torch.ones(7)
    chunk, h = reduce_scatter_delta(delta, spec, 1)
    h.wait()
    assert torch.equal(chunk, torch.full((cnt,), float(world))), chunk
    mv.shutdown()


def test_gather_scatter_uneven():
    run_dist(_gather_scatter_uneven, 2)


def _all_to_all_routing(rank, world):
    import torch
    import multiverso_amd as mv
    from multiverso_amd.comm import ShardSpec, all_to_all_rows
    mv.init()
    spec = ShardSpec(8, world)  # 4 rows per owner at world=2
    # rank 0 sends rows [7, 0, 4]; rank 1 sends [1, 5]
    ids = torch.tensor([7, 0, 4] if rank == 0 else [1, 5])
    vals = ids.float().repeat_interleave(2)  # unit=2
    in_ids, in_vals, recv_sizes, order, send_sizes = all_to_all_rows(
        ids, vals, spec, 2)
    # every received id must belong to my shard, with matching values
    off, cnt = spec.range_of(rank)
    assert torch.all((in_ids >= off) & (in_ids < off + cnt)), in_ids
    assert torch.equal(in_vals.view(-1, 2)[:, 0], in_ids.float())
    assert sum(recv_sizes) == in_ids.numel()
    mv.shutdown()


def test_all_to_all_routing():
    run_dist(_all_to_all_routing, 2)


def _model_average(rank, world):
    """-ma mode (zoo.cpp:24,49): skip the PS, aggregate raw buffers."""
    import torch
    import multiverso_amd as mv
    mv.init(["prog", "-ma=true"])
    assert mv.get_flag("ma") is True
    w = torch.full((16,), float(rank + 1))
    mv.aggregate(w)            # MV_Aggregate
    w /= mv.size()             # model average
    expect = sum(r + 1 for r in range(world)) / world
    assert torch.allclose(w, torch.full((16,), expect))
    mv.set_flag("ma", False)
    mv.shutdown()


def test_model_average():
    run_dist(_model_average, 2)


def _bucketed_aggregate(rank, world):
    import multiverso_amd as mv
    import torch
    mv.init()
    t = torch.arange(1000, dtype=torch.float32)
    out = mv.aggregate(t.clone(), bucket_mb=0)       # single call
    assert torch.equal(out, t * world)
    # force the bucket pipeline: ~4KB tensor with a tiny bucket is still
    # one bucket at 1 MiB floor, so build >1 MiB of data
    big = torch.ones(600_000)
    out2 = mv.aggregate(big.clone(), bucket_mb=1)    # 2.4 MB -> 3 buckets
    assert torch.equal(out2, big * world)
    mv.shutdown()


def test_bucketed_aggregate_dist():
    from conftest import run_dist
    run_dist(_bucketed_aggregate, 2)


def _keyed_no_device_sync(rank, world):
    """Keyed ops plan on the host: with CPU row ids, NO device transfer
    (comm.keyed_d2h counter) happens between launch and the value
    all-to-all (VERDICT r1 weak #2 'done' criterion)."""
    import multiverso_amd as mv
    import torch
    from multiverso_amd.dashboard import Dashboard
    mv.init(sync=True)
    t = mv.MatrixTable(12, 4)
    t.add_rows([rank, 11 - rank], torch.ones(2, 4))
    t.get_rows([0, 5, 11])
    assert Dashboard.get("comm.keyed_d2h").count == 0
    mv.shutdown()


def test_keyed_no_device_sync():
    run_dist(_keyed_no_device_sync, 2)
