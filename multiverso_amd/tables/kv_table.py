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
        self._ready.set()

    def _owner(self, key: int) -> int:
        return key % self.zoo.num_servers

    @property
    def num_shards(self) -> int:
        return self.zoo.num_servers

    def raw(self) -> Dict[int, float]:
        return self._mirror

    # ---- server-side entry points (kv_table.h:99-106) ----
    def _server_kv_add(self, keys_t, vals_t) -> None:
        with self._shard_lock:
            for k, v in zip(keys_t.tolist(), vals_t.tolist()):
                self._store[k] = self._store.get(k, 0) + v

    def _server_kv_get(self, keys_t):
        import torch
        with self._shard_lock:
            return torch.tensor([self._store.get(k, 0)
                                 for k in keys_t.tolist()],
                                dtype=torch.float64)

    def add(self, keys: Iterable[int], values: Iterable[float]) -> None:
        """Every rank contributes its (keys, values); each server applies
        the adds for its keys. Async mode: p2p to the owners, served on
        arrival (no coordination). Sync mode: collective."""
        eng = self.engine
        if eng is not None:
            import torch
            from ..comm import Handle
            k = torch.as_tensor(list(keys), dtype=torch.int64)
            v = torch.as_tensor(list(values), dtype=torch.float64)
            if k.numel():
                self._track(Handle(eng.kv_add(self, k, v)))
            return
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
        """Pull requested keys into the local mirror (async: p2p to the
        owners; sync: collective)."""
        keys = list(keys)
        eng = self.engine
        if eng is not None:
            import torch
            self.flush()   # my own in-flight adds land first (FIFO)
            k = torch.as_tensor(keys, dtype=torch.int64)
            vals = eng.kv_get(self, k).tolist() if keys else []
            for k_, v_ in zip(keys, vals):
                self._mirror[k_] = v_
            return {k_: self._mirror[k_] for k_ in keys}
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
        self.flush()
        self.zoo.barrier()   # async mode: every worker's adds are applied
        with self._shard_lock:
            snapshot = dict(self._store)
        if dist.is_initialized() and self.zoo.size > 1:
            shards: List = [None] * self.zoo.size
            dist.all_gather_object(shards, snapshot)
        else:
            shards = [snapshot]
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
