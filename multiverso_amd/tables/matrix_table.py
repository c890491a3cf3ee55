"""MatrixTable — 2-D dense table, row-sharded, whole-table or row-keyed ops.

Capability parity with the reference MatrixTable / unified Matrix table
(src/table/matrix_table.cpp, src/table/matrix.cpp): row sharding with
``num_row/num_servers`` rows per server and the remainder on the last
(matrix_table.cpp:24-45), whole-table Get/Add (:66-75, :387-417), row-set
Get/Add by ids (:59-147), optional uniform random init (float, :372-384),
raw-bytes Store/Load (:457-464).

MI355X mapping:
- whole-table Get = all-gather; whole-table Add = reduce-scatter + one
  fused updater kernel (K1-K4) on the owned rows.
- row-keyed ops = all-to-all exchange of (ids, values); the owner runs the
  K5/K6 gather/scatter kernels on its HBM shard. This replaces the
  per-server Request_Get/Add message fan-out (matrix_table.cpp:235-314)
  with per-destination bucketed traffic matched to xGMI's 7 p2p links.

Sparse/stale-aware Get (SparseMatrixTable's per-(worker,row) freshness
bitmap) is implemented in sparse_matrix.py on top of this class.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..comm import (Handle, ShardSpec, all_to_all_rows, all_to_all_values,
                    allgather_shards, reduce_scatter_delta)
from ..dashboard import monitor
from ..log import CHECK
from ..updaters import AddOption
from .base import Table


class MatrixTable(Table):
    def __init__(self, num_row: int, num_col: int,
                 dtype: torch.dtype = torch.float32,
                 updater_type: Optional[str] = None,
                 random_init: Optional[Tuple[float, float]] = None) -> None:
        super().__init__(updater_type)
        CHECK(num_row >= self.zoo.num_servers,
              f"num_row {num_row} must be >= num_servers")
        self.num_row = num_row
        self.num_col = num_col
        self.dtype = dtype
        self.spec = ShardSpec(num_row, self.zoo.num_servers)
        if self.zoo.is_server:
            self.row_offset, self.local_rows = self.spec.range_of(
                self.zoo.server_id)
        else:
            self.row_offset, self.local_rows = 0, 0  # ps_role=worker
        self.shard = torch.zeros(self.local_rows, num_col, dtype=dtype,
                                 device=self.device)
        if random_init is not None:
            # reference matrix_table.cpp:372-384: uniform [min, max), float,
            # identical content on every rank's view of its own rows.
            lo, hi = random_init
            g = torch.Generator(device="cpu").manual_seed(0x5EED + self.table_id)
            full = torch.rand(num_row, num_col, generator=g) * (hi - lo) + lo
            self.shard.copy_(full[self.row_offset:
                                  self.row_offset + self.local_rows])
        self._make_updater(self.shard.view(-1))
        # single-rank Add-deferral: an Add immediately followed by a
        # whole-table Get fuses into one updater kernel
        # (updater.update_and_copy) that writes both the shard and the
        # Get buffer, saving the Get's shard re-read (0.5 GB/step on the
        # headline config). Any other table op materializes the Add
        # first (flush).
        self._deferred = None  # (delta, option, delta._version)
        # subclasses with extra server state (SparseMatrixTable's
        # bitmap) publish readiness themselves after that state exists
        if not getattr(type(self), "_defer_ready", False):
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

    # ---- whole-table ops ----
    def _engine_get(self, eng, out, async_op):
        """Async-mode whole-table Get (worker.cpp:30-51 semantics)."""
        CHECK(self.zoo.is_worker, "ps_role=server ranks issue no Gets")
        self.flush()
        user_out = out
        if out is None or not out.is_contiguous():
            out = torch.empty(self.num_row, self.num_col, dtype=self.dtype,
                              device=self.device)
        CHECK(out.numel() == self.num_row * self.num_col,
              "Get buffer size mismatch")
        with monitor("worker.get"):
            pr = eng.whole_get(self, out.view(-1), self.num_col)

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

    def get(self, out: Optional[torch.Tensor] = None, async_op: bool = False):
        eng = self.engine
        if eng is not None:
            return self._engine_get(eng, out, async_op)
        d = self._deferred
        if d is not None and not async_op:
            self._deferred = None
            self._check_deferred(d)
            if out is None:
                out = torch.empty(self.num_row, self.num_col,
                                  dtype=self.dtype, device=self.device)
            CHECK(out.numel() == self.num_row * self.num_col,
                  "Get buffer size mismatch")
            # fused update+copy kernel needs a GPU fp32 contiguous target
            # (the deferral itself only happens on GPU shards); any other
            # out materializes the Add and falls through to the copy path
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
            out = torch.empty(self.num_row, self.num_col, dtype=self.dtype,
                              device=self.device)
        CHECK(out.numel() == self.num_row * self.num_col,
              "Get buffer size mismatch")
        if not out.is_contiguous():
            # gather needs a flat view; stage and copy into the user's
            # strided buffer afterwards
            out = torch.empty(self.num_row, self.num_col, dtype=self.dtype,
                              device=self.device)
        with monitor("worker.get"):
            h = allgather_shards(out.view(-1), self.shard.view(-1), self.spec,
                                 self.num_col, async_op=async_op)
        if out is not user_out and user_out is not None:
            if async_op:
                h.wait()
            user_out.copy_(out.view_as(user_out))
            out = user_out
        if async_op:
            self._track(h)
            return out, h
        return out

    def add(self, delta: torch.Tensor, option: Optional[AddOption] = None,
            async_op: bool = False) -> Handle:
        """Whole-table Add. Single-rank GPU adds DEFER until the next
        table op (fusing with an immediately following Get); every public
        read applies the pending add first, and in-place mutation of
        ``delta`` before then is a loud error. Use ``flush()`` to force
        materialization (e.g. before timing the update itself)."""
        CHECK(delta.numel() == self.num_row * self.num_col,
              "Add delta size mismatch")
        delta = delta.to(self.device, self.dtype).contiguous().view(-1)
        eng = self.engine
        if eng is not None:
            CHECK(self.zoo.is_worker, "ps_role=server ranks issue no Adds")
            self.flush()
            with monitor("worker.add"):
                h = Handle(eng.whole_add(self, delta, self.num_col, option,
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
            chunk, h = reduce_scatter_delta(delta, self.spec, self.num_col,
                                            async_op=async_op)
            if async_op:
                upd, opt = self.updater, option

                def epilogue() -> None:
                    with monitor("server.update"):
                        upd.update(chunk, opt)

                return self._track(Handle(h, epilogue))
            with monitor("server.update"):
                self.updater.update(chunk, option)
            return Handle()

    # ---- row-keyed ops (matrix_table.cpp:59-147, K5/K6) ----
    def _local_rows_of(self, ids: torch.Tensor) -> torch.Tensor:
        return ids - self.row_offset

    def _gather_local(self, local_ids: torch.Tensor) -> torch.Tensor:
        """K6 on the owned shard (HIP kernel for the f32 hot path;
        other dtypes use torch indexing — parity, not headline)."""
        if self.shard.is_cuda and self.dtype == torch.float32:
            from .. import ops
            return ops.module(required=True).row_gather(self.shard, local_ids)
        return self.shard[local_ids]

    def _scatter_update_local(self, local_ids: torch.Tensor,
                              vals: torch.Tensor,
                              option: Optional[AddOption],
                              assume_unique: bool = False) -> None:
        """K5 on the owned shard. default adds, sgd subtracts (the per-row
        updater dispatch of matrix_table.cpp:406-412); adagrad applies the
        keyed K15 form (duplicate rows: GPU races benignly via atomics,
        CPU pre-aggregates duplicates — both keep the G accumulate)."""
        from ..configure import get_flag
        if (get_flag("deterministic") and not assume_unique
                and local_ids.numel()):
            # fixed reduction order: pre-sum duplicate rows (sorted
            # segments) and scatter without atomics
            order = torch.argsort(local_ids, stable=True)
            sids = local_ids[order]
            v = vals.view(-1, self.num_col)[order]
            uids, counts = torch.unique_consecutive(sids,
                                                    return_counts=True)
            # fp64 cumsum-diff segmented sum: fixed order, no atomics
            cs = torch.zeros(v.size(0) + 1, self.num_col,
                             dtype=torch.float64, device=v.device)
            torch.cumsum(v.to(torch.float64), 0, out=cs[1:])
            ends = counts.cumsum(0)
            agg = (cs[ends] - cs[ends - counts]).to(v.dtype)
            local_ids, vals = uids, agg.reshape(-1)
            assume_unique = True
        if self.updater_type == "adagrad":
            from ..updaters import AddOption as _AO
            opt = option or _AO()
            gsq = self.updater.g_sqr.view(self.local_rows, self.num_col)
            vals2 = vals.view(-1, self.num_col)
            if self.shard.is_cuda and self.dtype == torch.float32:
                from .. import ops
                ops.module(required=True).row_scatter_adagrad(
                    self.shard, gsq, local_ids, vals2.contiguous(),
                    opt.learning_rate, opt.rho, self.updater.EPS,
                    assume_unique)
            else:
                urows, inv = torch.unique(local_ids, return_inverse=True)
                agg = torch.zeros(urows.numel(), self.num_col,
                                  dtype=self.dtype)
                agg.index_add_(0, inv, vals2)
                g = agg / opt.learning_rate
                gsq[urows] += g * g
                self.shard[urows] -= (opt.rho * g /
                                      torch.sqrt(gsq[urows]
                                                 + self.updater.EPS))
            return
        sign = {"default": 1.0, "sgd": -1.0}.get(self.updater_type)
        CHECK(sign is not None,
              f"row-keyed Add with updater '{self.updater_type}' is not "
              "supported yet (stateful updaters need segmented row update)")
        if self.shard.is_cuda and self.dtype == torch.float32:
            from .. import ops
            ops.module(required=True).row_scatter_add(
                self.shard, local_ids, vals.contiguous(), sign,
                assume_unique)
        else:
            self.shard.index_add_(0, local_ids,
                                  vals.view(-1, self.num_col) * sign)

    def get_rows(self, row_ids) -> torch.Tensor:
        """Row-subset Get: returns [len(row_ids), num_col] in caller order.

        ``row_ids`` is planned on the host (CPU ids are sync-free; device
        ids cost one D2H) and the split sizes ride the gloo control lane —
        no device sync between launch and the value all-to-all."""
        self.flush()
        eng = self.engine
        if eng is not None:
            CHECK(self.zoo.is_worker, "ps_role=server ranks issue no Gets")
            ids = torch.as_tensor(row_ids, dtype=torch.int64).cpu()
            with monitor("worker.get_rows"):
                return eng.keyed_get(self, ids, self.num_col)
        ids = torch.as_tensor(row_ids, dtype=torch.int64)
        with monitor("worker.get_rows"):
            in_ids, _, recv_sizes, order, send_sizes = all_to_all_rows(
                ids, None, self.spec, self.num_col, device=self.device)
            # serve: gather requested rows from my shard
            local = self._local_rows_of(in_ids)
            served = self._gather_local(local)
            # reply: route rows back to the requesters (what I received I
            # serve back — sizes swap roles, nothing recomputed)
            if self.zoo.size > 1:
                flat = all_to_all_values(served.view(-1), recv_sizes,
                                         send_sizes, self.num_col)
                got = flat.view(-1, self.num_col)
            else:
                # single rank: ids were never owner-sorted (order is the
                # identity), so the gather IS caller order — the
                # permutation write below would be a pure 31 us/chunk
                # identity index_put (measured, profiles r2 smaxprof2)
                return served.view(-1, self.num_col)
            # got is in owner-grouped order == order of sorted ids; undo sort
            out = torch.empty(ids.numel(), self.num_col, dtype=self.dtype,
                              device=self.device)
            out[order] = got
        return out

    def add_rows(self, row_ids, values: torch.Tensor,
                 option: Optional[AddOption] = None,
                 assume_unique: bool = False) -> None:
        """Row-subset Add (matrix_table.cpp:265-309 partition path).
        ``assume_unique=True`` asserts the caller's row_ids have no
        duplicates (e.g. a sorted-unique union) — at world size 1 the
        scatter then skips atomics; with multiple ranks incoming ids can
        still collide across ranks, so atomics stay."""
        self.flush()   # a deferred whole-table Add must land first
        ids = torch.as_tensor(row_ids, dtype=torch.int64)
        vals = values.to(self.device, self.dtype).contiguous()
        CHECK(vals.numel() == ids.numel() * self.num_col,
              "add_rows values size mismatch")
        eng = self.engine
        if eng is not None:
            CHECK(self.zoo.is_worker, "ps_role=server ranks issue no Adds")
            with monitor("worker.add_rows"):
                self._track(Handle(eng.keyed_add(
                    self, ids.cpu(), vals, self.num_col, option)))
            return
        with monitor("worker.add_rows"):
            in_ids, in_vals, _, _, _ = all_to_all_rows(
                ids, vals.view(-1), self.spec, self.num_col,
                device=self.device)
            if in_ids.numel():
                local = self._local_rows_of(in_ids)
                with monitor("server.update_rows"):
                    self._scatter_update_local(
                        local, in_vals.view(-1, self.num_col), option,
                        assume_unique=assume_unique and self.zoo.size == 1)

    # ---- server-side keyed entry points (async engine + local path) ----
    def _server_add_rows(self, local_ids: torch.Tensor,
                         vals2d: torch.Tensor, option) -> None:
        with self._shard_lock:
            with monitor("server.update_rows"):
                self._scatter_update_local(local_ids, vals2d, option)

    def _server_get_rows(self, local_ids: torch.Tensor) -> torch.Tensor:
        with self._shard_lock:
            return self._gather_local(local_ids)

    # ---- checkpoint (matrix_table.cpp:457-464) ----
    def store(self, path: str) -> None:
        """Raw whole-table bytes (row-major), streamed: each rank pwrites
        its own row range at its byte offset — no rank-0 staging."""
        self.flush()
        self._store_shard_stream(path, self.shard.view(-1),
                                 self.row_offset * self.num_col,
                                 self.num_row * self.num_col)

    def load(self, path: str) -> None:
        self.flush()
        self._load_shard_stream(path, self.shard.view(-1),
                                self.row_offset * self.num_col,
                                self.num_row * self.num_col)
