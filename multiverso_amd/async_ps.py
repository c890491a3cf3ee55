"""True asynchronous parameter-server execution (the reference's DEFAULT
mode).

Reference semantics being reproduced: a worker's Get/Add is served by the
owning server WHENEVER IT ARRIVES, with no global ordering and no
coordination between workers (src/server.cpp:36-58 dispatches each
Request_Get/Add on receipt; src/worker.cpp:30-76 partitions a request
across servers and waits only for its own replies). Workers never block
on each other; a straggler slows nobody. ``-sync=true`` selects the BSP
collective plane instead (tables' collective paths), exactly as the
reference's SyncServer replaces Server.

MI355X-native design: RCCL is a collective library — it has no
receive-from-any-source, so arrival-order service CANNOT ride the xGMI
data lane. The async mode therefore runs on the HOST CONTROL PLANE: each
rank that hosts shards runs one dedicated server thread (the reference's
server actor, actor.cpp:38-50) blocking on an any-source recv over a
gloo process group; payloads are host-staged and moved to/from the HBM
shard around the updater kernels. This is the latency/independence lane;
bulk synchronous traffic (bench, sync mode) stays on RCCL collectives.
Two disjoint groups keep the lanes race-free:

- ``req``: workers send requests; only server threads recv (any-source).
- ``rep``: server threads send replies; only worker callers recv.

gloo guarantees pairwise FIFO, so "my Get sees my prior Adds to the same
server" holds exactly as in the reference (same-worker-same-server
ordering; nothing more — matching server.cpp's arrival order). Replies
are matched by posting the reply recv at request-send time, so the
posting order always equals the server's per-pair processing order.

``ps_role`` worker/server splits (reference zoo.cpp:23,29-35): shards
live only on SERVER ranks; WORKER ranks address them remotely. Role ALL
(the default) makes every rank both, like the reference.
"""

from __future__ import annotations

import datetime
import threading
from typing import List, Optional

import torch
import torch.distributed as dist

from .dashboard import monitor
from .log import CHECK, log
from .updaters import AddOption

# ops (the Message MsgType equivalent, include/multiverso/message.h:13-24)
OP_ADD = 1          # whole-shard add: payload = my delta slice for you
OP_GET = 2          # whole-shard get: reply = your shard bytes
OP_ADD_ROWS = 3     # keyed add: keys + row values
OP_GET_ROWS = 4     # keyed get: keys; reply = row values
OP_KV_ADD = 5       # KV: int64 keys + f64 value deltas
OP_KV_GET = 6       # KV: int64 keys; reply = f64 values
OP_FLUSH = 7        # fence: ack after all my prior requests are applied
OP_FINISH = 8       # worker is done (Zoo StopPS / FinishTrain parity)
OP_GET_STALE = 9    # sparse table: rows stale FOR THIS WORKER; reply =
                    # count + global ids + values (matrix.cpp:478,540)

_HDR = 6  # int64 fields: [op, table_id, n_keys, n_vals, want_ack, has_opt]


def _opt_tensor(option: Optional[AddOption]) -> Optional[torch.Tensor]:
    if option is None:
        return None
    return torch.tensor([float(option.worker_id), option.momentum,
                         option.learning_rate, option.rho, option.lambda_],
                        dtype=torch.float64)


def _opt_from(t: torch.Tensor) -> AddOption:
    w, m, lr, rho, lam = t.tolist()
    return AddOption(int(w), m, lr, rho, lam)


class PendingReply:
    """A reply recv posted at request-send time (keeps send buffers alive
    until the transport is done with them)."""

    def __init__(self, works: List, refs: List, finish=None) -> None:
        self.works = works
        self.refs = refs
        self.finish = finish   # callable run once after all works complete
        self._done = False

    def wait(self) -> None:
        if self._done:
            return
        for w in self.works:
            w.wait()
        if self.finish is not None:
            self.finish()
        self.refs = []
        self._done = True


class AsyncEngine:
    def __init__(self, zoo) -> None:
        self.zoo = zoo
        tmo = datetime.timedelta(hours=24)  # server threads idle legally
        self.req = dist.new_group(backend="gloo", timeout=tmo)
        self.rep = dist.new_group(backend="gloo", timeout=tmo)
        self._thread: Optional[threading.Thread] = None
        self._server_error: Optional[BaseException] = None
        if zoo.is_server:
            self._thread = threading.Thread(target=self._serve_loop,
                                            name="mv-server", daemon=True)
            self._thread.start()

    # ------------------------------------------------------------------
    # worker side
    # ------------------------------------------------------------------
    # Request wire format: a fixed-size header message + AT MOST ONE
    # payload blob message per request (each gloo message costs ~0.2 ms
    # of latency on the host lane — round-2 measurement in
    # docs/ENGINEERING_NOTES.md — so an Add is 2 messages, not 3-4).
    # Blob layout, 8-byte aligned by construction:
    #   [opt: 5 f64 if has_opt][keys: n_keys i64][vals: n_vals dtype]
    def _send_request(self, dst: int, hdr_fields: List[int],
                      payloads: List[torch.Tensor]):
        """Returns (works, refs): pending isends + the buffers that must
        outlive them."""
        hdr = torch.tensor(hdr_fields, dtype=torch.int64)
        works = [dist.isend(hdr, dst, group=self.req)]
        refs: List[torch.Tensor] = [hdr]
        if payloads:
            if len(payloads) == 1:
                blob = payloads[0].contiguous().view(torch.uint8).view(-1)
            else:
                blob = torch.cat([p.contiguous().view(torch.uint8).view(-1)
                                  for p in payloads])
            works.append(dist.isend(blob, dst, group=self.req))
            refs.append(blob)
        return works, refs

    def whole_add(self, table, delta_flat: torch.Tensor, unit: int,
                  option: Optional[AddOption], want_ack: bool):
        """Partition my delta across servers; each server applies its
        slice on arrival (worker.cpp:53-76 -> server.cpp:48-58)."""
        zoo = self.zoo
        opt = _opt_tensor(option)
        works: List = []
        refs: List = []
        acks: List[torch.Tensor] = []
        for s in range(table.spec.n):
            off, cnt = table.spec.range_of(s)
            dst = zoo.server_ranks[s]
            piece = delta_flat[off * unit:(off + cnt) * unit]
            if dst == zoo.rank:
                table._server_apply_chunk(piece, option)
                continue
            payload = piece.cpu().contiguous()
            hdr = [OP_ADD, table.table_id, 0, payload.numel(),
                   1 if want_ack else 0, 0 if opt is None else 1]
            ps = ([] if opt is None else [opt]) + [payload]
            w, r = self._send_request(dst, hdr, ps)
            works += w
            refs += r
            if want_ack:
                ack = torch.empty(1, dtype=torch.int64)
                works.append(dist.irecv(ack, dst, group=self.rep))
                acks.append(ack)
        return PendingReply(works, refs + acks)

    def whole_get(self, table, out_flat: torch.Tensor, unit: int):
        """Request every server's shard; assemble replies into out.
        Returns a PendingReply (wait() completes the copy-in)."""
        zoo = self.zoo
        works: List = []
        refs: List = []
        slots = []   # (cpu_buf, dst_slice)
        for s in range(table.spec.n):
            off, cnt = table.spec.range_of(s)
            dst = zoo.server_ranks[s]
            dst_slice = out_flat[off * unit:(off + cnt) * unit]
            if dst == zoo.rank:
                table._server_read_chunk_into(dst_slice)
                continue
            hdr = [OP_GET, table.table_id, 0, cnt * unit, 0, 0]
            w, r = self._send_request(dst, hdr, [])
            works += w
            refs += r
            buf = torch.empty(cnt * unit, dtype=table.dtype)
            works.append(dist.irecv(buf, dst, group=self.rep))
            slots.append((buf, dst_slice))
            refs.append(buf)

        def finish() -> None:
            for buf, dst_slice in slots:
                dst_slice.copy_(buf.to(dst_slice.device))
        return PendingReply(works, refs, finish)

    def _plan_keyed(self, table, ids_cpu: torch.Tensor):
        base = max(table.spec.total // table.spec.n, 1)
        owners = torch.div(ids_cpu, base,
                           rounding_mode="floor").clamp_(max=table.spec.n - 1)
        order = torch.argsort(owners, stable=True)
        counts = torch.bincount(owners, minlength=table.spec.n).tolist()
        return order, counts

    def keyed_add(self, table, ids_cpu: torch.Tensor, vals: torch.Tensor,
                  unit: int, option: Optional[AddOption]):
        zoo = self.zoo
        order, counts = self._plan_keyed(table, ids_cpu)
        sorted_ids = ids_cpu[order]
        sorted_vals = vals.reshape(-1, unit)[order.to(vals.device)]
        opt = _opt_tensor(option)
        works: List = []
        refs: List = []
        off = 0
        for s, cnt in enumerate(counts):
            if cnt == 0:
                continue
            dst = zoo.server_ranks[s]
            keys = sorted_ids[off:off + cnt]
            piece = sorted_vals[off:off + cnt]
            off += cnt
            if dst == zoo.rank:
                table._server_add_rows(
                    (keys - table.row_offset).to(table.device),
                    piece.to(table.device), option)
                continue
            payload = piece.reshape(-1).cpu().contiguous()
            kcpu = keys.contiguous()
            hdr = [OP_ADD_ROWS, table.table_id, cnt, payload.numel(), 0,
                   0 if opt is None else 1]
            ps = ([] if opt is None else [opt]) + [kcpu, payload]
            w, r = self._send_request(dst, hdr, ps)
            works += w
            refs += r
        return PendingReply(works, refs)

    def keyed_get(self, table, ids_cpu: torch.Tensor,
                  unit: int) -> torch.Tensor:
        """Row-subset Get served per owning server on arrival; returns
        rows in caller order (synchronous — the caller needs the data)."""
        zoo = self.zoo
        order, counts = self._plan_keyed(table, ids_cpu)
        sorted_ids = ids_cpu[order]
        out = torch.empty(ids_cpu.numel(), unit, dtype=table.dtype,
                          device=table.device)
        works: List = []
        refs: List = []
        slots = []
        off = 0
        order_dev = order.to(table.device)
        for s, cnt in enumerate(counts):
            if cnt == 0:
                continue
            dst = zoo.server_ranks[s]
            keys = sorted_ids[off:off + cnt]
            rows = order_dev[off:off + cnt]
            off += cnt
            if dst == zoo.rank:
                got = table._server_get_rows(
                    (keys - table.row_offset).to(table.device))
                out[rows] = got
                continue
            kcpu = keys.contiguous()
            hdr = [OP_GET_ROWS, table.table_id, cnt, cnt * unit, 0, 0]
            w, r = self._send_request(dst, hdr, [kcpu])
            works += w
            refs += r
            buf = torch.empty(cnt * unit, dtype=table.dtype)
            works.append(dist.irecv(buf, dst, group=self.rep))
            slots.append((buf, rows))
            refs.append(buf)
        for w in works:
            w.wait()
        for buf, rows in slots:
            out[rows] = buf.view(-1, unit).to(table.device)
        return out

    def kv_add(self, table, keys: torch.Tensor, vals: torch.Tensor):
        """KVTable add: key % num_servers sharding (kv_table.h:49)."""
        zoo = self.zoo
        n = table.num_shards
        works: List = []
        refs: List = []
        for s in range(n):
            m = (keys % n) == s
            if not bool(m.any()):
                continue
            dst = zoo.server_ranks[s]
            k, v = keys[m].contiguous(), vals[m].contiguous()
            if dst == zoo.rank:
                table._server_kv_add(k, v)
                continue
            hdr = [OP_KV_ADD, table.table_id, k.numel(), v.numel(), 0, 0]
            w, r = self._send_request(dst, hdr, [k, v])
            works += w
            refs += r
        return PendingReply(works, refs)

    def kv_get(self, table, keys: torch.Tensor) -> torch.Tensor:
        zoo = self.zoo
        n = table.num_shards
        out = torch.zeros(keys.numel(), dtype=torch.float64)
        works: List = []
        refs: List = []
        slots = []
        for s in range(n):
            m = (keys % n) == s
            if not bool(m.any()):
                continue
            dst = zoo.server_ranks[s]
            k = keys[m].contiguous()
            if dst == zoo.rank:
                out[m] = table._server_kv_get(k)
                continue
            hdr = [OP_KV_GET, table.table_id, k.numel(), k.numel(), 0, 0]
            w, r = self._send_request(dst, hdr, [k])
            works += w
            refs += r
            buf = torch.empty(k.numel(), dtype=torch.float64)
            works.append(dist.irecv(buf, dst, group=self.rep))
            slots.append((buf, m))
        for w in works:
            w.wait()
        for buf, m in slots:
            out[m] = buf
        return out

    def sparse_get_stale(self, table, cache: torch.Tensor) -> int:
        """Stale-aware whole-table Get under async (the reference's
        sparse table ran under its async server the same way,
        matrix.cpp:461-478): each owner serves the rows stale for THIS
        worker on arrival and marks them fresh. Reply sizes are unknown
        to the requester, so the reply is count-then-payload on the FIFO
        pair (blocking recvs in send order). Returns rows received."""
        zoo = self.zoo
        unit = table.num_col
        me = zoo.worker_id
        total = 0
        remotes = []
        for s in range(table.spec.n):
            dst = zoo.server_ranks[s]
            if dst == zoo.rank:
                ids, vals = table._server_stale_for(me)
                if ids.numel():
                    cache[ids] = vals.to(cache.device)
                    total += ids.numel()
                continue
            hdr = [OP_GET_STALE, table.table_id, 0, 0, 0, 0]
            w, _r = self._send_request(dst, hdr, [])
            for ww in w:
                ww.wait()
            remotes.append(dst)
        elem = torch.empty(0, dtype=table.dtype).element_size()
        for dst in remotes:
            cnt = torch.empty(1, dtype=torch.int64)
            dist.recv(cnt, dst, group=self.rep)
            k = int(cnt[0])
            if not k:
                continue
            # count, then ONE [ids][vals] blob (matching the request
            # wire format; ids first keeps the vals view 8-byte aligned)
            blob = torch.empty(k * 8 + k * unit * elem, dtype=torch.uint8)
            dist.recv(blob, dst, group=self.rep)
            ids = blob[:k * 8].view(torch.int64)
            vals = blob[k * 8:].view(table.dtype)
            cache[ids] = vals.view(k, unit).to(cache.device)
            total += k
        return total

    # ---- fences ----
    def drain(self) -> None:
        """Fence: after this, every request THIS worker issued has been
        applied by its server (FIFO per pair + FLUSH ack). The Zoo
        barrier runs this first, so barrier keeps its reference meaning
        (all work issued before the barrier is visible after it)."""
        if self._server_error is not None:
            raise self._server_error
        if not self.zoo.is_worker:
            return
        works: List = []
        acks = []
        for dst in self.zoo.server_ranks:
            if dst == self.zoo.rank:
                continue
            hdr = torch.tensor([OP_FLUSH, 0, 0, 0, 1, 0], dtype=torch.int64)
            works.append(dist.isend(hdr, dst, group=self.req))
            ack = torch.empty(1, dtype=torch.int64)
            works.append(dist.irecv(ack, dst, group=self.rep))
            acks.append(ack)
        for w in works:
            w.wait()

    def shutdown(self) -> None:
        """FinishTrain (zoo.cpp:152-161). Two phases so no server thread
        can block forever: (1) drain + a global barrier — after it, NO
        worker will issue another request; (2) every worker sends FINISH
        to each remote server, so the only messages still in flight are
        exactly the FINISHes each server thread is waiting for."""
        self.drain()
        dist.barrier()
        if self.zoo.is_worker:
            for dst in self.zoo.server_ranks:
                if dst == self.zoo.rank:
                    continue
                hdr = torch.tensor([OP_FINISH, 0, 0, 0, 0, 0],
                                   dtype=torch.int64)
                dist.send(hdr, dst, group=self.req)
        if self._thread is not None:
            self._thread.join(timeout=300)
            CHECK(not self._thread.is_alive(),
                  "async server thread failed to drain at shutdown")
        if self._server_error is not None:
            raise self._server_error

    # ------------------------------------------------------------------
    # server side (the reference's server actor thread, server.cpp:36-58)
    # ------------------------------------------------------------------
    def _table(self, tid: int):
        """Resolve a table id, waiting out the creation race: requests
        are served on arrival, and a fast worker can legally send before
        this rank's constructor finished (the reference's table
        registration round-trip hid the same window)."""
        import time as _time
        deadline = _time.monotonic() + 120.0
        t = self.zoo._tables.get(tid)
        while t is None and _time.monotonic() < deadline:
            _time.sleep(0.001)
            t = self.zoo._tables.get(tid)
        CHECK(t is not None, f"async request for unknown table {tid}")
        CHECK(t._ready.wait(timeout=120.0),
              f"table {tid} never finished construction")
        return t

    def _serve_loop(self) -> None:
        try:
            zoo = self.zoo
            # local worker ops bypass the transport (gloo has no
            # loopback), so the thread only ever hears remote workers
            remote_workers = set(zoo.worker_ranks) - {zoo.rank}
            finished: set = set()
            while finished < remote_workers:
                hdr = torch.empty(_HDR, dtype=torch.int64)
                src = dist.recv(hdr, src=None, group=self.req)
                op, tid, n_keys, n_vals, want_ack, has_opt = hdr.tolist()
                if op == OP_FINISH:
                    finished.add(src)
                    continue
                if op == OP_FLUSH:
                    dist.send(torch.zeros(1, dtype=torch.int64), src,
                              group=self.rep)
                    continue
                self._serve_one(src, op, tid, n_keys, n_vals,
                                bool(want_ack), bool(has_opt))
        except BaseException as e:  # fail fast, loudly (CHECK parity)
            self._server_error = e
            log.error(f"async server thread died: {type(e).__name__}: {e}")
            raise

    def _serve_one(self, src: int, op: int, tid: int, n_keys: int,
                   n_vals: int, want_ack: bool, has_opt: bool) -> None:
        table = self._table(tid)
        # One blob recv mirrors the worker's one blob send (layout in
        # _send_request: [opt][keys][vals], 8-byte aligned sections)
        vals_dtype = None
        if n_vals and op in (OP_ADD, OP_ADD_ROWS):
            vals_dtype = table.dtype
        elif n_vals and op == OP_KV_ADD:
            vals_dtype = torch.float64
        vals_bytes = (0 if vals_dtype is None
                      else n_vals * torch.empty(0, dtype=vals_dtype)
                                         .element_size())
        nbytes = (40 if has_opt else 0) + n_keys * 8 + vals_bytes
        keys = None
        vals = None
        option = None
        if nbytes:
            blob = torch.empty(nbytes, dtype=torch.uint8)
            dist.recv(blob, src, group=self.req)
            off = 0
            if has_opt:
                option = _opt_from(blob[:40].view(torch.float64))
                off = 40
            if n_keys:
                keys = blob[off:off + n_keys * 8].view(torch.int64)
                off += n_keys * 8
            if vals_dtype is not None:
                vals = blob[off:off + vals_bytes].view(vals_dtype)

        if op == OP_ADD:
            with monitor("server.process_add"):
                table._server_apply_chunk(vals.to(table.device), option)
            if want_ack:
                dist.send(torch.zeros(1, dtype=torch.int64), src,
                          group=self.rep)
        elif op == OP_GET:
            with monitor("server.process_get"):
                out = torch.empty(n_vals, dtype=table.dtype)
                table._server_read_chunk_into(out)
            dist.send(out, src, group=self.rep)
        elif op == OP_ADD_ROWS:
            with monitor("server.process_add"):
                local = (keys - table.row_offset).to(table.device)
                table._server_add_rows(
                    local, vals.view(n_keys, -1).to(table.device), option)
        elif op == OP_GET_ROWS:
            with monitor("server.process_get"):
                local = (keys - table.row_offset).to(table.device)
                got = table._server_get_rows(local)
            dist.send(got.reshape(-1).cpu().contiguous(), src,
                      group=self.rep)
        elif op == OP_KV_ADD:
            table._server_kv_add(keys, vals)
        elif op == OP_KV_GET:
            dist.send(table._server_kv_get(keys), src, group=self.rep)
        elif op == OP_GET_STALE:
            worker = self.zoo.worker_ranks.index(src)
            with monitor("server.process_get"):
                ids, served = table._server_stale_for(worker)
            dist.send(torch.tensor([ids.numel()], dtype=torch.int64), src,
                      group=self.rep)
            if ids.numel():
                blob = torch.cat(
                    [ids.cpu().contiguous().view(torch.uint8).view(-1),
                     served.reshape(-1).cpu().contiguous()
                           .view(torch.uint8).view(-1)])
                dist.send(blob, src, group=self.rep)
        else:
            CHECK(False, f"unknown async op {op}")
