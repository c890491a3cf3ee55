"""Checkpoint/restore driver.

The reference ships per-table ``Serializable::Store/Load(Stream*)``
(table_interface.h:61-65) but the driver that called them was removed
(the Dockerfile still references checkpoint/restore test
targets). This module supplies the missing driver: ``checkpoint()`` /
``restore()`` walk every live table and stream each one's bytes through
the io layer. Per-table byte layout is identical to the reference's
(raw shard bytes in server order for Array/Matrix; see each table's
store/load)."""

from __future__ import annotations

import json
import os
from typing import Dict

from .log import CHECK, log
from .tables.base import _server_tables
from .zoo import Zoo


def checkpoint(directory: str) -> None:
    """MV_Checkpoint: write every registered table under ``directory``.
    Rank 0 writes a manifest + one file per table (table_<id>.bin)."""
    zoo = Zoo.get()
    CHECK(zoo.started, "checkpoint() requires an initialized runtime")
    if zoo.rank == 0:
        os.makedirs(directory, exist_ok=True)
    zoo.barrier()
    manifest: Dict[str, Dict] = {}
    for t in _server_tables:
        path = os.path.join(directory, f"table_{t.table_id}.bin")
        t.store(path)
        manifest[str(t.table_id)] = {
            "type": type(t).__name__,
            "file": os.path.basename(path),
        }
    if zoo.rank == 0:
        with open(os.path.join(directory, "manifest.json"), "w") as f:
            json.dump(manifest, f, indent=2)
    zoo.barrier()
    log.info(f"checkpointed {len(manifest)} tables to {directory}")


def restore(directory: str) -> None:
    """MV_Restore: load every registered table (created in the same order
    as at checkpoint time) from ``directory``."""
    zoo = Zoo.get()
    CHECK(zoo.started, "restore() requires an initialized runtime")
    with open(os.path.join(directory, "manifest.json")) as f:
        manifest = json.load(f)
    for t in _server_tables:
        entry = manifest.get(str(t.table_id))
        if entry is None:
            log.error(f"no checkpoint entry for table {t.table_id}; skipped")
            continue
        CHECK(entry["type"] == type(t).__name__,
              f"table {t.table_id} type mismatch: checkpoint has "
              f"{entry['type']}, live table is {type(t).__name__}")
        t.load(os.path.join(directory, entry["file"]))
    zoo.barrier()
    log.info(f"restored {len(manifest)} tables from {directory}")
