"""Tests for io streams, checkpoint driver, ASyncBuffer, SparseFilter,
and the stale-aware SparseMatrixTable (single- and multi-process)."""

import time

import pytest
import torch

import multiverso_amd as mv
from conftest import run_dist


@pytest.fixture()
def env():
    mv.init()
    yield
    mv.shutdown()


# ---- io ----

def test_local_stream_roundtrip(tmp_path):
    p = f"file://{tmp_path}/x.bin"
    with mv.StreamFactory.get_stream(p, "w") as s:
        s.write(b"hello\nworld\n")
    with mv.StreamFactory.get_stream(p, "r") as s:
        assert s.read() == b"hello\nworld\n"
    r = mv.TextReader(p)
    assert r.get_line() == "hello"
    assert r.get_line() == "world"
    assert r.get_line() is None


def test_uri_parse():
    u = mv.URI("hdfs://nn/path/f")
    assert u.scheme == "hdfs" and u.path == "nn/path/f"
    assert mv.URI("/plain/path").scheme == "file"
    with pytest.raises(NotImplementedError):
        mv.StreamFactory.get_stream("hdfs://x/y", "r")
    with pytest.raises(ValueError):
        mv.StreamFactory.get_stream("s3://x/y", "r")


# ---- checkpoint driver ----

def test_checkpoint_restore_driver(env, tmp_path):
    a = mv.ArrayTable(16)
    m = mv.MatrixTable(4, 4)
    a.add(torch.arange(16, dtype=torch.float32))
    m.add(torch.ones(4, 4))
    mv.checkpoint(str(tmp_path / "ckpt"))
    mv.shutdown()
    mv.init()
    a2 = mv.ArrayTable(16)
    m2 = mv.MatrixTable(4, 4)
    mv.restore(str(tmp_path / "ckpt"))
    assert torch.equal(a2.get(), torch.arange(16, dtype=torch.float32))
    assert torch.equal(m2.get(), torch.ones(4, 4))


# ---- ASyncBuffer ----

def test_async_buffer_prefetch():
    calls = []

    def fill(buf):
        time.sleep(0.01)
        buf.append(len(calls))
        calls.append(1)

    ab = mv.ASyncBuffer([], [], fill)
    b0 = ab.get()
    assert b0 == [0]
    b1 = ab.get()
    assert b1 == [1]
    b0b = ab.get()
    assert b0b[0] == 0 and len(b0b) == 2  # same buffer object, refilled


# ---- sparse filter ----

def test_sparse_filter_roundtrip():
    from multiverso_amd import sparse_filter as sf
    v = torch.zeros(1000)
    v[[3, 500, 999]] = torch.tensor([1.0, -2.0, 3.5])
    payload, comp = sf.filter_in(v)
    assert comp and payload.numel() == 1 + 6
    assert torch.equal(sf.filter_out(payload, comp, 1000), v)
    dense = torch.ones(10)
    payload, comp = sf.filter_in(dense)
    assert not comp and torch.equal(payload, dense)


def test_sparse_filter_exact_above_2pow24():
    """Indices are int32 bit patterns, not float values: payloads larger
    than 2^24 elements (where float32 integers stop being exact) must
    round-trip exactly (float-valued indices silently
    corrupted rows past 16.7M elements)."""
    from multiverso_amd import sparse_filter as sf
    n = (1 << 24) + 8
    v = torch.zeros(n)
    hot = [0, (1 << 24) - 1, (1 << 24), (1 << 24) + 1, n - 1]
    v[hot] = torch.tensor([1.0, 2.0, 3.0, 4.0, 5.0])
    payload, comp = sf.filter_in(v)
    assert comp
    back = sf.filter_out(payload, comp, n)
    assert torch.equal(back[hot], v[hot])
    assert float(back.sum()) == 15.0   # nothing landed anywhere else


# ---- sparse matrix table ----

def test_sparse_matrix_local(env):
    t = mv.SparseMatrixTable(8, 2)
    cache = torch.full((8, 2), -1.0)
    assert t.get_into(cache) == 8      # everything stale initially
    assert torch.equal(cache, torch.zeros(8, 2))
    assert t.get_into(cache) == 0      # now fresh
    t.add_rows([2, 6], torch.ones(2, 2))
    got = t.get_into(cache)
    assert got == 2
    assert torch.equal(cache[2], torch.ones(2))
    assert torch.equal(cache[0], torch.zeros(2))


def _sparse_matrix_dist(rank, world):
    import multiverso_amd as mv
    import torch
    mv.init(sync=True)
    t = mv.SparseMatrixTable(9, 3)
    cache = torch.zeros(9, 3)
    n = t.get_into(cache)
    assert n == 9, n
    assert t.get_into(cache) == 0
    # rank 0 adds row 1; both ranks then see it stale and re-pull
    if rank == 0:
        t.add_rows([1], torch.ones(1, 3))
    else:
        t.add_rows([], torch.zeros(0, 3))
    n = t.get_into(cache)
    assert n == 1, (rank, n)
    assert torch.equal(cache[1], torch.ones(3))
    mv.shutdown()


def test_sparse_matrix_dist():
    run_dist(_sparse_matrix_dist, 2)


def _checkpoint_dist(rank, world):
    import multiverso_amd as mv
    import torch
    import tempfile, os
    mv.init(sync=True)
    d = os.path.join(tempfile.gettempdir(), "mv_ckpt_test")
    a = mv.ArrayTable(10)
    a.add(torch.ones(10))
    mv.checkpoint(d)
    mv.shutdown()
    mv.init(sync=True)
    a2 = mv.ArrayTable(10)
    mv.restore(d)
    got = a2.get()
    assert torch.equal(got, torch.full((10,), float(world))), got
    mv.shutdown()


def test_checkpoint_dist():
    run_dist(_checkpoint_dist, 2)


def _streaming_checkpoint(rank, world, tmpdir):
    """Streaming sharded store: every rank pwrites only its own shard in
    bounded chunks (chunk bound shrunk below one shard here), and the
    resulting file is byte-identical to the whole table row-major — the
    reference layout. No rank-0 full-table staging exists to fall back
    on, so this also proves a table bigger than any single staging
    buffer checkpoints correctly."""
    import os
    import numpy as np
    import torch
    import multiverso_amd as mv
    from multiverso_amd.tables.base import Table
    mv.init(sync=True)
    Table._IO_CHUNK_BYTES = 16   # 4 floats per chunk -> many chunks/shard
    t = mv.MatrixTable(9, 3)     # uneven shards: 4 + 5 rows
    full = torch.arange(27, dtype=torch.float32).view(9, 3)
    t.add(full / world)
    path = os.path.join(tmpdir, "m.bin")
    t.store(path)
    raw = np.fromfile(path, dtype=np.float32)
    assert np.allclose(raw.reshape(9, 3), full.numpy()), raw
    # wipe, then streamed load restores every shard
    t.shard.zero_()
    t.load(path)
    assert torch.allclose(t.get(), full)
    mv.shutdown()


def test_streaming_checkpoint_dist(tmp_path):
    import functools
    run_dist(functools.partial(_streaming_checkpoint,
                               tmpdir=str(tmp_path)), 2)


def _sparse_filter_wire(rank, world):
    """Stale-row replies travel SparseFilter-compressed (the initial pull
    of an all-zeros table is ~free) and decompress exactly."""
    import multiverso_amd as mv
    import torch
    mv.init(sync=True)
    from multiverso_amd.dashboard import Dashboard
    t = mv.SparseMatrixTable(10, 4)
    assert t.use_sparse_filter
    cache = torch.full((10, 4), -1.0)
    n = t.get_into(cache)   # all-zero table: payloads compress to ~nothing
    assert n == 10 and torch.equal(cache, torch.zeros(10, 4))
    if rank == 0:
        t.add_rows([2], torch.tensor([[1.0, 0.0, 0.0, 2.0]]))
    else:
        t.add_rows([], torch.zeros(0, 4))
    n = t.get_into(cache)
    assert n == 1, n
    assert torch.equal(cache[2], torch.tensor([1.0, 0.0, 0.0, 2.0]))
    bytes_in = Dashboard.get("sparse_filter.bytes_in").elapsed_ms
    bytes_out = Dashboard.get("sparse_filter.bytes_out").elapsed_ms
    assert bytes_in > 0 and bytes_out < bytes_in  # compression happened
    mv.shutdown()


def test_sparse_filter_wire_dist():
    from conftest import run_dist
    run_dist(_sparse_filter_wire, 2)


def _sparse_prefetch_dist(rank, world):
    """Pipelined stale get (the unified Matrix's is_pipeline,
    matrix.cpp:384-420): prefetch overlaps with an add; rows marked
    fresh at issue are re-invalidated by the interleaved add, so the
    NEXT get re-pulls them."""
    import multiverso_amd as mv
    import torch
    mv.init(sync=True)
    t = mv.SparseMatrixTable(9, 3)
    cache = torch.zeros(9, 3)
    h = t.prefetch_into(cache)      # issue the initial full pull
    # interleaved keyed add while the exchange is in flight
    if rank == 0:
        t.add_rows([1], torch.ones(1, 3))
    else:
        t.add_rows([], torch.zeros(0, 3))
    h.wait()
    assert h.rows == 9
    # prefetch snapshotted PRE-add values; the add re-invalidated row 1
    n = t.get_into(cache)
    assert n == 1, n
    assert torch.equal(cache[1], torch.ones(3))
    assert t.get_into(cache) == 0
    mv.shutdown()


def test_sparse_prefetch_dist():
    run_dist(_sparse_prefetch_dist, 2)


def test_sparse_prefetch_local(env):
    import torch
    t = mv.SparseMatrixTable(6, 2)
    cache = torch.zeros(6, 2)
    h = t.prefetch_into(cache)
    h.wait()
    assert h.rows == 6
    assert t.get_into(cache) == 0
