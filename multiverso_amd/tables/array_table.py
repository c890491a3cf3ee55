"""ArrayTable — 1-D dense table, contiguous-sharded across ranks.

Capability parity with the reference ArrayTable
(src/table/array_table.cpp, include/multiverso/table/array_table.h):
whole-table Get/Add only (the reference always sends key −1), contiguous
partition ``size/num_servers`` with the remainder on the last server
(array_table.cpp:11-21), server-side updater on Add (:116-127), Access
copy-out on Get (:130-141), raw-bytes Store/Load (:144-151).

MI355X mapping: Get = all-gather of shards into the caller's buffer;
Add = reduce-scatter of the delta followed by one fused updater kernel on
the owned shard. Both can be issued async and overlap with compute.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..comm import Handle, ShardSpec, allgather_shards, reduce_scatter_delta
from ..dashboard import monitor
from ..log import CHECK
from ..updaters import AddOption
from .base import Table


class ArrayTable(Table):
    def __init__(self, size: int, dtype: torch.dtype = torch.float32,
                 updater_type: Optional[str] = None) -> None:
        super().__init__(updater_type)
        CHECK(size >= self.zoo.num_servers,
              f"table size {size} must be >= num_servers "
              f"{self.zoo.num_servers} (reference array_table.cpp:14)")
        self.size = size
        self.dtype = dtype
        self.spec = ShardSpec(size, self.zoo.num_servers)
        if self.zoo.is_server:
            off, cnt = self.spec.range_of(self.zoo.server_id)
        else:
            cnt = 0   # ps_role=worker: no shard hosted here
        self.shard = torch.zeros(cnt, dtype=dtype, device=self.device)
        self.row_offset = 0   # keyed-op base (arrays have no keyed ops)
        self._make_updater(self.shard)
        # single-rank Add deferral + Add/Get fusion — same mechanism and
        # semantics as MatrixTable (see matrix_table.py)
        self._deferred = None  # (delta, option, delta._version)
        self._ready.set()

    def _check_deferred(self, d) -> None:
        CHECK(d[0]._version == d[2],
              "the delta tensor passed to Add was mutated in place before "
              "the Add was applied (single-GPU deferred-Add fast path "
              "snapshots at the next table op; pass a fresh tensor)")

    def flush(self) -> None:
        d = self._deferred
        if d is not None:
            self._deferred = None
            self._check_deferred(d)
            with monitor("server.update"):
                self.updater.update(d[0], d[1])
        super().flush()

    # ---- worker ops ----
    def get(self, out: Optional[torch.Tensor] = None,
            async_op: bool = False):
        """Whole-table Get (array_table.cpp:24-66 semantics)."""
        eng = self.engine
        if eng is not None:
            return self._engine_get(eng, out, async_op)
        d = self._deferred
        if d is not None and not async_op:
            self._deferred = None
            self._check_deferred(d)
            if out is None:
                out = torch.empty(self.size, dtype=self.dtype,
                                  device=self.device)
            CHECK(out.numel() == self.size, "Get buffer size mismatch")
            # fused update+copy kernel needs a GPU fp32 contiguous target
            # (see matrix_table.get); anything else materializes the Add
            if (out.is_contiguous() and out.is_cuda
                    and out.dtype == self.dtype):
                with monitor("server.update"):
                    self.updater.update_and_copy(d[0], d[1], out.view(-1))
                return out
            with monitor("server.update"):
                self.updater.update(d[0], d[1])
        self.flush()
        user_out = out
        if out is None:
            out = torch.empty(self.size, dtype=self.dtype, device=self.device)
        CHECK(out.numel() == self.size, "Get buffer size mismatch")
        if not out.is_contiguous():
            out = torch.empty(self.size, dtype=self.dtype, device=self.device)
        with monitor("worker.get"):
            h = allgather_shards(out.view(-1), self.shard, self.spec, 1,
                                 async_op=async_op)
        if out is not user_out and user_out is not None:
            if async_op:
                h.wait()
            user_out.copy_(out.view_as(user_out))
            out = user_out
        if async_op:
            self._track(h)
            return out, h
        return out

    def _engine_get(self, eng, out, async_op):
        """Async-mode Get: request each server's shard; served on
        arrival (worker.cpp:30-51 -> server.cpp:36-46)."""
        CHECK(self.zoo.is_worker, "ps_role=server ranks issue no Gets")
        self.flush()
        user_out = out
        if out is None or not out.is_contiguous():
            out = torch.empty(self.size, dtype=self.dtype,
                              device=self.device)
        CHECK(out.numel() == self.size, "Get buffer size mismatch")
        with monitor("worker.get"):
            pr = eng.whole_get(self, out.view(-1), 1)

        def _finish() -> None:
            if user_out is not None and user_out is not out:
                user_out.copy_(out.view_as(user_out))
        h = Handle(pr, _finish)
        ret = user_out if user_out is not None else out
        if async_op:
            self._track(h)
            return ret, h
        h.wait()
        return ret

    def add(self, delta: torch.Tensor, option: Optional[AddOption] = None,
            async_op: bool = False) -> Handle:
        """Whole-table Add: reduce-scatter + updater (server.cpp:48 →
        array_table.cpp:116-127); in async mode, per-server slices
        applied on arrival."""
        CHECK(delta.numel() == self.size, "Add delta size mismatch")
        delta = delta.to(self.device, self.dtype).contiguous().view(-1)
        eng = self.engine
        if eng is not None:
            CHECK(self.zoo.is_worker, "ps_role=server ranks issue no Adds")
            self.flush()
            with monitor("worker.add"):
                h = Handle(eng.whole_add(self, delta, 1, option,
                                         want_ack=not async_op))
            if async_op:
                return self._track(h)
            h.wait()
            return h
        if self.zoo.size == 1 and self.shard.is_cuda:
            self.flush()                 # at most one deferred Add
            self._deferred = (delta, option, delta._version)
            return Handle()
        with monitor("worker.add"):
            chunk, h = reduce_scatter_delta(delta, self.spec, 1,
                                            async_op=async_op)
            if async_op:
                upd, opt = self.updater, option

                def epilogue() -> None:
                    with monitor("server.update"):
                        upd.update(chunk, opt)

                # inner Handle resolves the collective; epilogue applies the
                # updater exactly once (Handle latches on _done).
                return self._track(Handle(h, epilogue))
            with monitor("server.update"):
                self.updater.update(chunk, option)
            return Handle()

    # ---- checkpoint (Store/Load, array_table.cpp:144-151) ----
    def store(self, path: str) -> None:
        """Write whole-table raw bytes, byte-identical to the reference's
        concatenated per-shard Store layout. Streaming: each rank pwrites
        its own shard at its offset — no rank-0 full-table gather."""
        self.flush()
        off, _ = self.spec.range_of(self.zoo.server_id)
        self._store_shard_stream(path, self.shard, off, self.size)

    def load(self, path: str) -> None:
        """Each rank reads only its own shard slice (streamed)."""
        self.flush()
        off, _ = self.spec.range_of(self.zoo.server_id)
        self._load_shard_stream(path, self.shard, off, self.size)
