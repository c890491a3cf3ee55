"""KVTable — distributed key→value map (counters / sparse scalars).

Capability parity with the reference KVTable
(include/multiverso/table/kv_table.h): worker keeps a local mirror
(``raw()``), keys partition to servers by ``key % num_servers``
(kv_table.h:49), server applies ``table_[k] += v`` (:99-106), Get pulls
the requested keys into the mirror.

This table carries small host-side counters (the WordEmbedding word-count
table), not tensors, so it lives on the CPU and exchanges
via object collectives — latency-bound control traffic, deliberately kept
off the xGMI data lane (tiny control messages go over the
host lane). Store/Load is implemented (the reference Fatal'd, kv_table.h:108).
"""

from __future__ import annotations

import pickle
from typing import Dict, Iterable, List

import torch.distributed as dist

from .base import Table


class KVTable(Table):
    def __init__(self) -> None:
        super().__init__(updater_type="default")
        self._store: Dict[int, float] = {}   # server-side shard (my keys)
        self._mirror: Dict[int, float] = {}  # worker-side local mirror

    def _owner(self, key: int) -> int:
        return key % self.zoo.num_servers

    def raw(self) -> Dict[int, float]:
        return self._mirror

    def add(self, keys: Iterable[int], values: Iterable[float]) -> None:
        """Collective: every rank contributes its (keys, values); each
        server applies the adds for its keys."""
        mine: List = list(zip(keys, values))
        if dist.is_initialized() and self.zoo.size > 1:
            gathered: List = [None] * self.zoo.size
            dist.all_gather_object(gathered, mine)
        else:
            gathered = [mine]
        for contrib in gathered:
            for k, v in contrib:
                if self._owner(k) == self.zoo.server_id:
                    self._store[k] = self._store.get(k, 0) + v

    def get(self, keys: Iterable[int]) -> Dict[int, float]:
        """Collective: pull requested keys into the local mirror."""
        keys = list(keys)
        if dist.is_initialized() and self.zoo.size > 1:
            # each server broadcasts its shard's answers for all requests
            wanted: List = [None] * self.zoo.size
            dist.all_gather_object(wanted, keys)
            union = set()
            for ks in wanted:
                union.update(ks)
            answers = {k: self._store[k] for k in union
                       if self._owner(k) == self.zoo.server_id
                       and k in self._store}
            all_answers: List = [None] * self.zoo.size
            dist.all_gather_object(all_answers, answers)
            merged: Dict[int, float] = {}
            for a in all_answers:
                merged.update(a)
        else:
            merged = dict(self._store)
        for k in keys:
            self._mirror[k] = merged.get(k, 0)
        return {k: self._mirror[k] for k in keys}

    # ---- checkpoint ----
    def store(self, path: str) -> None:
        if dist.is_initialized() and self.zoo.size > 1:
            shards: List = [None] * self.zoo.size
            dist.all_gather_object(shards, self._store)
        else:
            shards = [self._store]
        if self.zoo.rank == 0:
            merged: Dict[int, float] = {}
            for s in shards:
                merged.update(s)
            with open(path, "wb") as f:
                pickle.dump(merged, f)
        self.zoo.barrier()

    def load(self, path: str) -> None:
        with open(path, "rb") as f:
            merged = pickle.load(f)
        self._store = {k: v for k, v in merged.items()
                       if self._owner(k) == self.zoo.server_id}
        self.zoo.barrier()
