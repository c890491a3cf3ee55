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
        self._pending: List[Handle] = []
        _server_tables.append(self)

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
