"""Zoo — the per-process runtime singleton.

Capability parity with the reference Zoo (src/zoo.cpp:41-188,
include/multiverso/zoo.h): system bring-up/teardown, rank↔role bookkeeping,
barrier, table registration, and the model-average aggregate entry point.

MI355X-native redesign: instead of MPI init plus a
rank-0 controller actor doing Control_Register round-trips, each process is
one rank of a ``torch.distributed`` process group — RCCL over xGMI on GPU
nodes (backend "nccl" IS RCCL on ROCm), gloo on CPU. Rendezvous (the
reference's RegisterNode / RegisterController, zoo.cpp:116-145 /
controller.cpp:46-72) is the torchrun env rendezvous; Barrier
(zoo.cpp:164-176) is ``dist.barrier``. Every rank is worker+server
(ps_role=default → Role::ALL, zoo.cpp:23,29-35); table shards live in that
rank's HBM. There is no message-passing actor chain on the data path at
all: Get/Add are collectives issued from the caller's thread onto a side
HIP stream (comm.py).
"""

from __future__ import annotations

import datetime
import os
import threading
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from .configure import get_flag, parse_cmd_flags
from .log import CHECK, log


class Role:
    NONE = 0
    WORKER = 1
    SERVER = 2
    ALL = 3


class Zoo:
    _instance: Optional["Zoo"] = None
    _lock = threading.Lock()

    def __init__(self) -> None:
        self.started = False
        self.rank = 0
        self.size = 1
        self.role = Role.ALL
        self.device: torch.device = torch.device("cpu")
        self._tables: Dict[int, object] = {}
        self._next_table_id = 0
        self._owns_pg = False
        self.backend = "none"
        # host control lane (tiny control messages ride a
        # CPU lane, not the xGMI data lane): a gloo subgroup that carries
        # CPU scalars/counts even when the main backend is RCCL. None at
        # world size 1 or when the main group is already gloo.
        self.control_pg = None
        # explicit rendezvous (MV_NetBind/MV_NetConnect,
        # src/multiverso.cpp:58-68): rank + rank->endpoint map set before
        # init, replacing the launcher-env rendezvous the way the
        # reference's ZMQ Bind/Connect replaced mpirun
        self._bound_rank: Optional[int] = None
        self._endpoints: Optional[List[str]] = None
        # async PS engine (the reference's DEFAULT mode, server.cpp:36-58):
        # active when sync=false and world>1 — table ops are served on
        # arrival by per-rank server threads instead of collectives
        self.async_engine = None
        # ps_role bookkeeping (zoo.cpp:23,29-35): which global ranks host
        # shards / issue worker ops; role ALL (default) -> all of them
        self.server_ranks: List[int] = [0]
        self.worker_ranks: List[int] = [0]

    # ---- explicit rendezvous (MV_NetBind / MV_NetConnect) ----
    def net_bind(self, rank: int, endpoint: str) -> bool:
        """Declare this process's rank (endpoint kept for the map;
        reference ZMQNetWrapper::Bind, zmq_net.h)."""
        CHECK(not self.started, "net_bind must precede init")
        self._bound_rank = rank
        return True

    def net_connect(self, ranks: List[int], endpoints: List[str]) -> bool:
        """Provide the full rank->endpoint map; rank 0's endpoint hosts
        the rendezvous store (reference ZMQNetWrapper::Connect)."""
        CHECK(not self.started, "net_connect must precede init")
        CHECK(len(ranks) == len(endpoints), "ranks/endpoints mismatch")
        eps = [""] * len(ranks)
        for r, e in zip(ranks, endpoints):
            eps[r] = e
        self._endpoints = eps
        return True

    # ---- singleton ----
    @classmethod
    def get(cls) -> "Zoo":
        with cls._lock:
            if cls._instance is None:
                cls._instance = Zoo()
            return cls._instance

    # ---- lifecycle ----
    def start(self, argv: Optional[List[str]] = None,
              backend: Optional[str] = None) -> List[str]:
        """MV_Init equivalent (multiverso.cpp:11, zoo.cpp:41-102)."""
        if self.started:
            return argv or []
        rest = parse_cmd_flags(argv or [])

        role = get_flag("ps_role")
        self.role = {"default": Role.ALL, "worker": Role.WORKER,
                     "server": Role.SERVER, "none": Role.NONE}.get(role, Role.ALL)
        CHECK(self.role in (Role.ALL, Role.WORKER, Role.SERVER),
              f"unsupported ps_role={role}")

        explicit = (self._bound_rank is not None
                    and self._endpoints is not None)
        if explicit:
            world_size = len(self._endpoints)
        else:
            world_size = int(os.environ.get("WORLD_SIZE", "1"))
        cuda = torch.cuda.is_available()

        if cuda:
            if explicit:
                local_rank = self._bound_rank % torch.cuda.device_count()
            else:
                local_rank = int(os.environ.get(
                    "LOCAL_RANK", os.environ.get("RANK", "0")))
            torch.cuda.set_device(local_rank)
            self.device = torch.device("cuda", local_rank)
        else:
            self.device = torch.device("cpu")

        if dist.is_initialized():
            self._owns_pg = False
            self.backend = dist.get_backend()
        elif explicit and world_size > 1:
            self.backend = backend or os.environ.get(
                "MV_BACKEND", "nccl" if cuda else "gloo")
            dist.init_process_group(
                backend=self.backend,
                init_method=f"tcp://{self._endpoints[0]}",
                rank=self._bound_rank, world_size=world_size,
                timeout=datetime.timedelta(seconds=300),
            )
            self._owns_pg = True
        elif world_size > 1 or "MASTER_ADDR" in os.environ:
            self.backend = backend or os.environ.get(
                "MV_BACKEND", "nccl" if cuda else "gloo")
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29517")
            dist.init_process_group(
                backend=self.backend,
                timeout=datetime.timedelta(seconds=300),
            )
            self._owns_pg = True
        else:
            self.backend = "none"  # single-process degenerate mode

        if dist.is_initialized():
            self.rank = dist.get_rank()
            self.size = dist.get_world_size()
            if self.size > 1 and self.backend == "nccl":
                # gloo control lane alongside RCCL: CPU control scalars
                # (uneven-stream participation flags, keyed-op split
                # sizes) must not ride the GPU data lane — RCCL rejects
                # CPU tensors and a count exchange on-device forces a
                # stream sync.
                self.control_pg = dist.new_group(backend="gloo")
        else:
            self.rank, self.size = 0, 1

        # role map (zoo.cpp:29-35): exchange every rank's role over the
        # host lane, derive the worker/server rank lists that shard specs
        # and the async engine address
        if self.size > 1:
            vec = torch.zeros(self.size, dtype=torch.int64)
            vec[self.rank] = self.role
            dist.all_reduce(vec, group=self.control_pg)
            roles = vec.tolist()
        else:
            roles = [self.role]
        self.server_ranks = [r for r, ro in enumerate(roles)
                             if ro & Role.SERVER]
        self.worker_ranks = [r for r, ro in enumerate(roles)
                             if ro & Role.WORKER]
        CHECK(len(self.server_ranks) > 0,
              "at least one rank must host shards (ps_role=server|default)")
        CHECK(len(self.worker_ranks) > 0,
              "at least one rank must be a worker (ps_role=worker|default)")
        split = (len(self.server_ranks) != self.size
                 or len(self.worker_ranks) != self.size)
        if self.size > 1 and not bool(get_flag("sync")):
            # the reference's default (async) mode: per-rank server
            # threads serve Get/Add on arrival; -sync=true selects the
            # BSP collective plane instead
            from .async_ps import AsyncEngine
            self.async_engine = AsyncEngine(self)
        CHECK(not split or self.async_engine is not None,
              "ps_role worker/server splits require the async PS mode "
              "(sync=false): the BSP collective plane needs every rank "
              "to be worker+server")

        lvl = str(get_flag("log_level")).lower()
        from .log import LogLevel
        if lvl in ("debug", "info", "error", "fatal"):
            log.reset_log_level(getattr(LogLevel, lvl.upper()))
        omp = int(get_flag("omp_threads"))
        if omp > 0:
            # reference updater.cpp:18-19 OpenMP thread count -> the CPU
            # fallback's intra-op pool; 0 (default) keeps torch's choice
            torch.set_num_threads(omp)

        self.started = True
        log.debug(f"Zoo started: rank {self.rank}/{self.size} "
                  f"backend={self.backend} device={self.device}")
        self.barrier()
        return rest

    def stop(self, finalize_net: bool = True) -> None:
        """MV_ShutDown equivalent (zoo.cpp:104-161). Best-effort after a
        FatalError: log.fatal tears the process groups down to unblock
        peers, so every dist call here may find them gone — shutdown
        must still leave the process exitable."""
        if not self.started:
            return
        try:
            if self.async_engine is not None:
                eng = self.async_engine
                eng.shutdown()     # drain + barrier + FinishTrain + join
                self.async_engine = None
                if dist.is_initialized():
                    dist.destroy_process_group(eng.req)
                    dist.destroy_process_group(eng.rep)
            else:
                self.barrier()
        except Exception as e:
            log.error(f"shutdown after comm teardown (best-effort): {e}")
            self.async_engine = None
        from .tables.base import free_tables
        free_tables()
        self._tables.clear()
        self._next_table_id = 0
        try:
            if dist.is_initialized() and self.control_pg is not None:
                dist.destroy_process_group(self.control_pg)
        except Exception:
            pass
        self.control_pg = None
        try:
            if self._owns_pg and dist.is_initialized() and finalize_net:
                dist.destroy_process_group()
        except Exception:
            pass
        self.started = False

    # ---- bookkeeping (zoo.h:19-85) ----
    @property
    def num_workers(self) -> int:
        return len(self.worker_ranks)

    @property
    def num_servers(self) -> int:
        return len(self.server_ranks)

    @property
    def worker_id(self) -> int:
        try:
            return self.worker_ranks.index(self.rank)
        except ValueError:
            return -1

    @property
    def server_id(self) -> int:
        try:
            return self.server_ranks.index(self.rank)
        except ValueError:
            return -1

    @property
    def is_worker(self) -> bool:
        return self.rank in self.worker_ranks

    @property
    def is_server(self) -> bool:
        return self.rank in self.server_ranks

    def register_table(self, table) -> int:
        tid = self._next_table_id
        self._next_table_id += 1
        self._tables[tid] = table
        return tid

    # ---- collectives ----
    def barrier(self) -> None:
        if self.async_engine is not None:
            # reference MV_Barrier meaning preserved under async: all
            # work issued before the barrier is visible after it
            self.async_engine.drain()
        if dist.is_initialized():
            if self.backend == "nccl":
                dist.barrier(device_ids=[self.device.index])
                torch.cuda.synchronize(self.device)
            else:
                dist.barrier()

    def aggregate(self, tensor: torch.Tensor,
                  bucket_mb: Optional[int] = None) -> torch.Tensor:
        """In-place sum-allreduce — MV_Aggregate (src/multiverso.cpp:53-56,
        net.cpp:27-41). On GPU this is rcclAllReduce over xGMI; payloads
        larger than ``bucket_mb`` (flag ``bucket_mb``) are issued as a
        pipeline of async bucket all-reduces — the rebuild's tunable for
        the reference's 4096-byte small-message switch
        (allreduce_engine.cpp:35), sized for 7 xGMI links instead."""
        if not dist.is_initialized():
            return tensor
        if (not tensor.is_cuda) and self.control_pg is not None:
            # CPU scalar/flag aggregates under an RCCL main backend ride
            # the gloo control lane (RCCL rejects CPU tensors)
            dist.all_reduce(tensor, op=dist.ReduceOp.SUM,
                            group=self.control_pg)
            return tensor
        if bucket_mb is None:
            bucket_mb = int(get_flag("bucket_mb"))
        nbytes = tensor.numel() * tensor.element_size()
        if bucket_mb <= 0 or nbytes <= bucket_mb * (1 << 20):
            dist.all_reduce(tensor, op=dist.ReduceOp.SUM)
            return tensor
        flat = tensor.view(-1)
        step = bucket_mb * (1 << 20) // tensor.element_size()
        works = []
        for off in range(0, flat.numel(), step):
            works.append(dist.all_reduce(flat[off:off + step],
                                         op=dist.ReduceOp.SUM,
                                         async_op=True))
        for w in works:
            w.wait()
        return tensor

    @property
    def sync_mode(self) -> bool:
        return bool(get_flag("sync"))
