"""CLI self-test dispatcher — the reference's `multiverso.test` binary
(Test/main.cpp:12-24: dispatch kv|array|net|matrix|allreduce, intended
as `mpirun -np 4 ./multiverso.test <name>`; the Docker image ran them,
deploy/docker/Dockerfile:104-110).

Launch (any world size; every rank runs the same oracle):

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \\
      --master-addr 127.0.0.1 -m multiverso_amd.selftest array

Single-process invocation (`python -m multiverso_amd.selftest kv`)
degenerates to 1 worker + 1 server through the same code paths, exactly
like the reference's boost unit tests.
"""

from __future__ import annotations

import sys

import torch

import multiverso_amd as mv


def test_array() -> None:
    """Test/test_array_table.cpp:11-47: sync-mode Add x3 / Get x3 with
    the world-scaled exact-value oracle."""
    mv.init(sync=True)
    t = mv.ArrayTable(100)
    for it in range(3):
        t.add(torch.full((100,), 2.0)).wait()
        got = t.get()
        expect = 2.0 * (it + 1) * mv.size()
        assert torch.equal(got, torch.full((100,), expect)), (it, got[0])
    mv.shutdown()


def test_kv() -> None:
    """Test/test_kv_table.cpp:8-33: KV add/get roundtrip (incl. negative
    values)."""
    mv.init(sync=True)
    t = mv.KVTable()
    t.add([1, 5, 9], [2.0, -3.0, 4.0])
    got = t.get([1, 5, 9, 77])
    w = mv.size()
    assert got[1] == 2.0 * w and got[5] == -3.0 * w and got[9] == 4.0 * w
    assert got[77] == 0.0
    mv.shutdown()


def test_net() -> None:
    """Test/test_net.cpp:9-95: raw transport send/recv between ranks.
    Here the transport is the async engine's p2p lanes: each worker's
    request/reply framing (header + multi-payload message) makes the
    round trip through a real server thread."""
    mv.init()   # async mode = the p2p message lanes
    t = mv.MatrixTable(8 * max(mv.size(), 2), 4)
    rows = [mv.rank(), mv.rank() + 8]
    t.add_rows(rows, torch.full((2, 4), float(mv.rank() + 1)))
    got = t.get_rows(rows)   # served by the owning rank's server thread
    assert got.shape == (2, 4)
    mv.barrier()
    mv.shutdown()


def test_matrix() -> None:
    """Test/test_matrix_table.cpp:9-107: dense whole-table + row-subset
    Add/Get with exact-value verification (iteration count reduced from
    the reference's 10k — same oracle per iteration)."""
    mv.init(sync=True)
    num_row, num_col = 30, 5
    t = mv.MatrixTable(num_row, num_col)
    w = mv.size()
    for it in range(20):
        t.add(torch.ones(num_row, num_col))
        got = t.get()
        assert torch.equal(
            got, torch.full((num_row, num_col), float((it + 1) * w))), it
    rows = [0, 7, 29]
    t.add_rows(rows, torch.ones(3, num_col))
    got = t.get_rows(rows)
    assert torch.equal(got, torch.full((3, num_col), float(21 * w)))
    mv.shutdown()


def test_allreduce() -> None:
    """Test/test_allreduce.cpp:10-21: `-ma` mode, MV_Aggregate of one
    int == world size."""
    mv.init(["-ma=true"])
    x = torch.ones(1, dtype=torch.int32)
    mv.aggregate(x)
    assert int(x[0]) == mv.size(), x
    mv.shutdown()


DISPATCH = {
    "array": test_array,
    "kv": test_kv,
    "net": test_net,
    "matrix": test_matrix,
    "allreduce": test_allreduce,
}


def main() -> int:
    if len(sys.argv) != 2 or sys.argv[1] not in DISPATCH:
        print(f"usage: python -m multiverso_amd.selftest "
              f"{'|'.join(DISPATCH)}", file=sys.stderr)
        return 2
    name = sys.argv[1]
    DISPATCH[name]()
    print(f"selftest {name}: PASS", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
