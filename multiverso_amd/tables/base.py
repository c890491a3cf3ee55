"""Table base: HBM-resident server shard + collective worker ops.

The reference splits every table into a WorkerTable client handle and a
ServerTable storage shard connected by the worker/server/communicator actor
chain (include/multiverso/table_interface.h:24-75).
In the MI355X rebuild every rank is worker+server, so one Table object owns
BOTH sides: the local shard (a torch tensor in this GPU's HBM — 288 GB per
GPU means even the reference's 6B-parameter word-embedding claim fits
8-way sharded) and the client ops, which are collectives
(comm.py) instead of messages.

Async Get/Add return a Handle (the reference's Waiter) and are tracked in
a per-table pending list; any synchronous op or ``flush`` resolves them
first, which preserves the reference ordering guarantee that a Get
observes every Add issued before it on this worker.
"""

from __future__ import annotations

import os
import threading
from typing import List, Optional

import torch

from ..comm import Handle
from ..log import CHECK
from ..updaters import AddOption, create_updater
from ..zoo import Zoo

_server_tables: List["Table"] = []


def free_tables() -> None:
    """table_factory::FreeServerTables equivalent (table_factory.cpp:11-16)."""
    _server_tables.clear()


class Table:
    def __init__(self, updater_type: Optional[str] = None) -> None:
        zoo = Zoo.get()
        CHECK(zoo.started, "multiverso_amd.init() must be called before "
                           "creating tables")
        self.zoo = zoo
        self.table_id = zoo.register_table(self)
        if updater_type is None:
            from ..configure import get_flag
            updater_type = get_flag("updater_type")
        self.updater_type = updater_type
        self.updater = None  # created by subclass once the shard exists
        # async mode: a fast worker's request can arrive before THIS
        # rank finished constructing the table (the reference's
        # RegisterTable round-trip subsumed this); the server thread
        # waits on _ready before touching the shard
        self._ready = threading.Event()
        self._pending: List[Handle] = []
        # serializes the async server thread against local worker ops on
        # the same shard (the reference's one-consumer-thread-per-actor
        # guarantee, actor.cpp:38-50, collapsed to a lock)
        self._shard_lock = threading.Lock()
        _server_tables.append(self)

    @property
    def engine(self):
        """The async PS engine, or None in sync/single-process mode."""
        return self.zoo.async_engine

    # ---- server-side entry points (ServerTable::ProcessAdd/ProcessGet,
    # table_interface.h:61-75) — called by the async server thread AND by
    # local worker fast paths; both hold the shard lock ----
    def _server_apply_chunk(self, chunk: torch.Tensor, option) -> None:
        with self._shard_lock:
            self.updater.update(
                chunk.to(self.updater.shard.device,
                         self.updater.shard.dtype), option)

    def _server_read_chunk_into(self, out_flat: torch.Tensor) -> None:
        with self._shard_lock:
            src = self.updater.shard.view(-1)
            if out_flat.device == src.device:
                out_flat.copy_(src)
            else:
                out_flat.copy_(src.cpu() if out_flat.device.type == "cpu"
                               else src.to(out_flat.device))

    def _make_updater(self, shard: torch.Tensor) -> None:
        self.updater = create_updater(self.updater_type, shard)

    # ---- async bookkeeping ----
    def _track(self, handle: Handle) -> Handle:
        self._pending.append(handle)
        return handle

    def flush(self) -> None:
        """Resolve all in-flight async ops for this table."""
        pending, self._pending = self._pending, []
        for h in pending:
            h.wait()

    # ---- helpers ----
    @property
    def device(self) -> torch.device:
        return self.zoo.device

    def default_option(self) -> AddOption:
        return AddOption(worker_id=self.zoo.worker_id)

    # ---- streaming sharded checkpoint I/O ----
    # Byte layout stays identical to the reference's per-shard Store
    # (array_table.cpp:144-151 / matrix_table.cpp:457-464 write raw shard
    # bytes in server order == the whole table row-major, since sharding
    # is contiguous). Unlike round 1 there is NO rank-0 full-table
    # staging: every rank pwrites its own shard at its byte offset in
    # bounded chunks, so a table larger than one rank's free memory
    # checkpoints fine.
    _IO_CHUNK_BYTES = (64 << 20)  # per-chunk host staging bound

    def _store_shard_stream(self, path: str, flat_shard: torch.Tensor,
                            elem_offset: int, total_elems: int) -> None:
        elem = flat_shard.element_size()
        if self.zoo.rank == 0:
            with open(path, "wb") as f:
                f.truncate(total_elems * elem)
        self.zoo.barrier()   # file exists at full size before any pwrite
        chunk_elems = max(self._IO_CHUNK_BYTES // elem, 1)
        n = flat_shard.numel()
        if n:
            with open(path, "r+b") as f:
                f.seek(elem_offset * elem)
                for off in range(0, n, chunk_elems):
                    piece = flat_shard[off:off + chunk_elems]
                    f.write(piece.cpu().numpy().tobytes())
        self.zoo.barrier()

    def _load_shard_stream(self, path: str, flat_shard: torch.Tensor,
                           elem_offset: int, total_elems: int) -> None:
        import numpy as np
        elem = flat_shard.element_size()
        CHECK(os.path.getsize(path) == total_elems * elem,
              f"checkpoint size mismatch: {path} has "
              f"{os.path.getsize(path)} bytes, table needs "
              f"{total_elems * elem}")
        chunk_elems = max(self._IO_CHUNK_BYTES // elem, 1)
        n = flat_shard.numel()
        np_dtype = np.dtype(str(flat_shard.dtype).replace("torch.", ""))
        if n:
            with open(path, "rb") as f:
                f.seek(elem_offset * elem)
                for off in range(0, n, chunk_elems):
                    cnt = min(chunk_elems, n - off)
                    buf = np.fromfile(f, dtype=np_dtype, count=cnt)
                    flat_shard[off:off + cnt].copy_(
                        torch.from_numpy(buf).to(flat_shard.device))
        self.zoo.barrier()
