"""multiverso_amd — MI355X-native parameter-server training framework.

A from-scratch rebuild of Microsoft Multiverso's capabilities for a single
8×MI355X node: one process per GPU, table traffic over RCCL/xGMI
collectives, server shards resident in HBM3E, server-side updaters as
hand-written CDNA4 HIP kernels. See SURVEY.md for the capability map.

Public API parity with the reference Python binding
(binding/python/multiverso/api.py, tables.py):

    import multiverso_amd as mv
    mv.init(sync=True)
    tbl = mv.ArrayTableHandler(1000, init_value=...)
    tbl.add(delta); v = tbl.get()
    mv.barrier(); mv.shutdown()
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np
import torch

from .configure import define_flag, get_flag, parse_cmd_flags, set_flag
from .dashboard import Dashboard, Monitor, Timer, monitor
from .log import CHECK, CHECK_NOTNULL, FatalError, LogLevel, log
from .async_buffer import ASyncBuffer
from .checkpoint import checkpoint, restore
from .io import Stream, StreamFactory, TextReader, URI
from .tables import ArrayTable, KVTable, MatrixTable, SparseMatrixTable
from .updaters import AddOption
from .zoo import Role, Zoo

__version__ = "0.1.0"


# ---------------------------------------------------------------------------
# Core API (reference api.py:12-75 / multiverso.h MV_* functions)
# ---------------------------------------------------------------------------

def init(args: Optional[List[str]] = None, sync: bool = False,
         backend: Optional[str] = None) -> None:
    """MV_Init. ``sync=True`` selects BSP semantics (reference api.py:12-34
    injects ``-sync=true``; here it sets the flag directly)."""
    if sync:
        set_flag("sync", True)
    Zoo.get().start(args if args is not None else [], backend=backend)


def shutdown(finalize_net: bool = True) -> None:
    """MV_ShutDown."""
    Zoo.get().stop(finalize_net)


def barrier() -> None:
    """MV_Barrier."""
    Zoo.get().barrier()


def net_bind(rank: int, endpoint: str) -> bool:
    """MV_NetBind (src/multiverso.cpp:58-62): declare this process's rank
    for an explicit, launcher-free rendezvous (the reference's ZMQ Bind
    deployment mode)."""
    return Zoo.get().net_bind(rank, endpoint)


def net_connect(ranks: List[int], endpoints: List[str]) -> bool:
    """MV_NetConnect (src/multiverso.cpp:64-68): full rank->endpoint map;
    rank 0's endpoint hosts the rendezvous store. Call before init()."""
    return Zoo.get().net_connect(ranks, endpoints)


def workers_num() -> int:
    return Zoo.get().num_workers


def servers_num() -> int:
    return Zoo.get().num_servers


def worker_id() -> int:
    return Zoo.get().worker_id


def server_id() -> int:
    return Zoo.get().server_id


def is_master_worker() -> bool:
    return worker_id() == 0


def rank() -> int:
    return Zoo.get().rank


def size() -> int:
    return Zoo.get().size


def aggregate(tensor: torch.Tensor,
              bucket_mb: Optional[int] = None) -> torch.Tensor:
    """MV_Aggregate — in-place sum-allreduce (model-average mode),
    bucket-pipelined above the ``bucket_mb`` threshold."""
    return Zoo.get().aggregate(tensor, bucket_mb=bucket_mb)


# ---------------------------------------------------------------------------
# Handler classes (reference binding/python/multiverso/tables.py)
# ---------------------------------------------------------------------------

def _to_tensor(data, device, num_col: Optional[int] = None) -> torch.Tensor:
    if isinstance(data, torch.Tensor):
        t = data.to(device=device, dtype=torch.float32)
    else:
        t = torch.as_tensor(np.asarray(data, dtype=np.float32), device=device)
    return t


class ArrayTableHandler:
    """reference tables.py:38-81: float32 1-D table; on creation the master
    worker adds ``init_value`` and the others add zeros (both sync) so the
    table starts at init_value exactly once."""

    def __init__(self, size: int, init_value=None) -> None:
        self._table = ArrayTable(size, torch.float32)
        self.size = size
        if init_value is not None:
            init = _to_tensor(init_value, self._table.device).reshape(-1)
            CHECK(init.numel() == size, "init_value size mismatch")
            if not is_master_worker():
                init = torch.zeros_like(init)
            self.add(init, sync=True)
            # async mode: the master's init add must be VISIBLE to every
            # worker before any of them proceeds (in sync mode the add is
            # already a collective; the barrier is then a cheap no-op
            # beyond the reference's own post-init barrier)
            barrier()

    def get(self) -> torch.Tensor:
        return self._table.get()

    def add(self, data, sync: bool = False,
            option: Optional[AddOption] = None) -> None:
        t = _to_tensor(data, self._table.device).reshape(-1)
        h = self._table.add(t, option=option, async_op=not sync)
        if sync:
            h.wait()


class MatrixTableHandler:
    """reference tables.py:84-165: float32 2-D table with whole-table and
    row-set get/add."""

    def __init__(self, num_row: int, num_col: int, init_value=None) -> None:
        self._table = MatrixTable(num_row, num_col, torch.float32)
        self.num_row, self.num_col = num_row, num_col
        if init_value is not None:
            init = _to_tensor(init_value, self._table.device)
            init = init.reshape(num_row, num_col) if init.numel() > 1 else \
                torch.full((num_row, num_col), float(init.item()),
                           device=self._table.device)
            if not is_master_worker():
                init = torch.zeros_like(init)
            self.add(init, sync=True)
            barrier()   # see ArrayTableHandler: init visible to all

    def get(self, row_ids: Optional[Sequence[int]] = None) -> torch.Tensor:
        if row_ids is None:
            return self._table.get()
        return self._table.get_rows(row_ids)

    def add(self, data, row_ids: Optional[Sequence[int]] = None,
            sync: bool = False, option: Optional[AddOption] = None) -> None:
        t = _to_tensor(data, self._table.device)
        if row_ids is None:
            h = self._table.add(t.reshape(-1), option=option,
                                async_op=not sync)
            if sync:
                h.wait()
        else:
            self._table.add_rows(row_ids, t.reshape(len(list(row_ids)),
                                                    self.num_col),
                                 option=option)


__all__ = [
    "init", "shutdown", "barrier", "net_bind", "net_connect",
    "workers_num", "servers_num", "worker_id",
    "server_id", "is_master_worker", "rank", "size", "aggregate",
    "ArrayTable", "MatrixTable", "SparseMatrixTable", "KVTable",
    "ArrayTableHandler", "MatrixTableHandler",
    "checkpoint", "restore", "ASyncBuffer",
    "Stream", "StreamFactory", "TextReader", "URI",
    "AddOption", "set_flag", "get_flag", "define_flag", "parse_cmd_flags",
    "Dashboard", "Monitor", "Timer", "monitor",
    "log", "LogLevel", "CHECK", "CHECK_NOTNULL", "FatalError",
    "Zoo", "Role",
]
