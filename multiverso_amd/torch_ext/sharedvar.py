"""MVSharedTensor — the torch generation of the Theano sharedvar binding.

Capability parity with binding/python/multiverso/theano_ext/sharedvar.py:
``MVSharedVariable`` wraps a shared tensor; ``mv_sync`` adds the local
value-delta to an ArrayTable and then sets the tensor to the table value
(:12-74); a module registry collects shared tensors and
``sync_all_mv_shared_vars`` (:77-99) syncs them all."""

from __future__ import annotations

from typing import List

import torch

from ..tables.array_table import ArrayTable


class MVSharedTensor:
    def __init__(self, tensor: torch.Tensor) -> None:
        self.tensor = tensor
        self._table = ArrayTable(tensor.numel())
        # publish initial value once (master adds value, others zeros —
        # tables.py:50-57 protocol)
        from .. import is_master_worker
        init = tensor.detach().reshape(-1).float()
        if not is_master_worker():
            init = torch.zeros_like(init)
        self._table.add(init)
        # async mode: the master's init must be visible to every worker
        # before anyone reads (no-op cost in sync mode — see handlers)
        from ..zoo import Zoo
        Zoo.get().barrier()
        self._last = self._table.get().clone()
        with torch.no_grad():
            self.tensor.reshape(-1).copy_(self._last.to(tensor.device))

    def mv_sync(self) -> None:
        """add(current - last_synced); then set to the server value."""
        cur = self.tensor.detach().reshape(-1).float().to(self._last.device)
        self._table.add(cur - self._last)
        got = self._table.get()
        self._last = got.clone()
        with torch.no_grad():
            self.tensor.reshape(-1).copy_(got.to(self.tensor.device))


_registry: List[MVSharedTensor] = []


def mv_shared(tensor: torch.Tensor) -> MVSharedTensor:
    sv = MVSharedTensor(tensor)
    _registry.append(sv)
    return sv


def sync_all_mv_shared() -> None:
    for sv in _registry:
        sv.mv_sync()


def clear_registry() -> None:
    _registry.clear()
