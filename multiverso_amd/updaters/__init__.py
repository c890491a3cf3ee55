"""Server-side updaters — each a single fused CDNA4 HIP kernel on GPU.

Capability parity with the reference updater family (selected by the
``updater_type`` flag, src/updater/updater.cpp:46-57):

- default: ``data += delta``            (updater.cpp:22-29, K1)
- sgd:     ``data -= delta``            (sgd_updater.h:14-19, K2 — the
           worker pre-scales delta by lr)
- momentum:``m = mu*m + (1-mu)*delta; data -= m``
           (momentum_updater.h:17-25, K3)
- adagrad: ``g = delta/lr; G += g*g; data -= rho * g / sqrt(G + eps)``
           (adagrad_updater.h:23-41, K4). The reference decrements G and
           copies the accumulator row by value so the state update is lost
           (SURVEY.md §2.3 flags this as a bug to fix) — we keep the
           intended accumulate semantics. In the collective data plane the
           reduce-scattered delta is the sum over workers, so there is one
           accumulator per shard rather than one per (worker, shard).
- dcasgd / dcasgda: delay-compensated ASGD (selected at updater.cpp:51-54
           from a submodule absent in the snapshot; math from Zheng et
           al., ICML 2017) with per-worker backup rows.

``AddOption`` keeps the reference's 20-byte wire envelope
(include/multiverso/updater/updater.h:10-70) for C-API parity.

Each updater's ``update`` is ONE pass over the shard: on GPU it dispatches
to the in-tree HIP extension (multiverso_amd.ops) — fused read-modify-write
kernels, float4-vectorized, grid-stride (memory-bound per the CDNA4 guide:
the bound is HBM3E bytes, so fusing m/G state updates into the same pass
halves traffic vs. composing torch ops). CPU fallback uses torch ops and
is numerically identical in fp32.
"""

from __future__ import annotations

import struct
from typing import Dict, Optional, Type

import torch


class AddOption:
    """20-byte (5 x 4B) add-option envelope; field order matches the
    reference union layout: worker_id(i), momentum(f), lr(f), rho(f),
    lambda(f)."""

    __slots__ = ("worker_id", "momentum", "learning_rate", "rho", "lambda_")

    def __init__(self, worker_id: int = 0, momentum: float = 0.0,
                 learning_rate: float = 0.01, rho: float = 0.1,
                 lambda_: float = 0.1) -> None:
        self.worker_id = worker_id
        self.momentum = momentum
        self.learning_rate = learning_rate
        self.rho = rho
        self.lambda_ = lambda_

    def to_bytes(self) -> bytes:
        return struct.pack("<iffff", self.worker_id, self.momentum,
                           self.learning_rate, self.rho, self.lambda_)

    @classmethod
    def from_bytes(cls, b: bytes) -> "AddOption":
        w, m, lr, rho, lam = struct.unpack("<iffff", b[:20])
        return cls(w, m, lr, rho, lam)


def _hip_ops():
    """In-tree HIP extension, or None on CPU-only hosts. On a GPU host a
    missing extension is a hard error (no silent eager fallback)."""
    from .. import ops
    return ops.module(required=torch.cuda.is_available())


class Updater:
    name = "default"

    def __init__(self, shard: torch.Tensor) -> None:
        self.shard = shard

    def update(self, delta: torch.Tensor, option: Optional[AddOption]) -> None:
        if self.shard.is_cuda:
            _hip_ops().add_inplace(self.shard, delta)
        else:
            self.shard.add_(delta)

    def update_and_copy(self, delta: torch.Tensor,
                        option: Optional[AddOption],
                        out: torch.Tensor) -> None:
        """Fused Add+Get (single-rank fast path): apply the update AND
        stream the updated shard into ``out`` in the same pass — saves
        the Get's shard re-read. Every updater has a fused HIP kernel;
        the CPU fallback composes update + copy."""
        if self.shard.is_cuda and self.shard.dtype in (torch.float32,
                                                        torch.float64):
            _hip_ops().sgd_copy_update(self.shard, delta, out, 1.0)
        else:
            self.update(delta, option)
            out.copy_(self.shard)

    def access(self, out: torch.Tensor) -> None:
        """K7: shard copy-out (updater.cpp:32-36)."""
        out.copy_(self.shard)


class SGDUpdater(Updater):
    name = "sgd"

    def update(self, delta: torch.Tensor, option: Optional[AddOption]) -> None:
        if self.shard.is_cuda:
            _hip_ops().sgd_update(self.shard, delta)
        else:
            self.shard.sub_(delta)

    def update_and_copy(self, delta, option, out) -> None:
        if self.shard.is_cuda and self.shard.dtype in (torch.float32,
                                                       torch.float64):
            _hip_ops().sgd_copy_update(self.shard, delta, out, -1.0)
        else:
            self.update(delta, option)
            out.copy_(self.shard)


class MomentumUpdater(Updater):
    name = "momentum"

    def __init__(self, shard: torch.Tensor) -> None:
        super().__init__(shard)
        self.smooth_gradient = torch.zeros_like(shard)

    def update(self, delta: torch.Tensor, option: Optional[AddOption]) -> None:
        mu = option.momentum if option else 0.0
        if self.shard.is_cuda:
            _hip_ops().momentum_update(self.shard, self.smooth_gradient,
                                       delta, float(mu))
        else:
            self.smooth_gradient.mul_(mu).add_(delta, alpha=1.0 - mu)
            self.shard.sub_(self.smooth_gradient)

    def update_and_copy(self, delta, option, out) -> None:
        if self.shard.is_cuda and self.shard.dtype == torch.float32:
            mu = option.momentum if option else 0.0
            _hip_ops().momentum_copy_update(self.shard, self.smooth_gradient,
                                            delta, out, float(mu))
        else:
            self.update(delta, option)
            out.copy_(self.shard)


class AdaGradUpdater(Updater):
    name = "adagrad"
    EPS = 1e-6

    def __init__(self, shard: torch.Tensor) -> None:
        super().__init__(shard)
        self.g_sqr = torch.zeros_like(shard)

    def update(self, delta: torch.Tensor, option: Optional[AddOption]) -> None:
        opt = option or AddOption()
        lr, rho = opt.learning_rate, opt.rho
        if self.shard.is_cuda:
            _hip_ops().adagrad_update(self.shard, self.g_sqr, delta,
                                      float(lr), float(rho), self.EPS)
        else:
            g = delta / lr
            self.g_sqr.add_(g * g)
            self.shard.sub_(rho * g / torch.sqrt(self.g_sqr + self.EPS))

    def update_and_copy(self, delta, option, out) -> None:
        if self.shard.is_cuda and self.shard.dtype == torch.float32:
            opt = option or AddOption()
            _hip_ops().adagrad_copy_update(self.shard, self.g_sqr, delta,
                                           out, float(opt.learning_rate),
                                           float(opt.rho), self.EPS)
        else:
            self.update(delta, option)
            out.copy_(self.shard)


class DCASGDUpdater(Updater):
    """DC-ASGD ("dcasgd", reference updater.cpp:51 — the updater itself
    lives in a submodule absent from the snapshot; math from Zheng et al.,
    "Asynchronous SGD with Delay Compensation", ICML 2017):

        w   -= lr * (g + lambda * g*g * (w - bak_k))
        bak_k = w

    ``bak_k`` is the per-worker backup of the weights worker k last
    synchronized (AddOption.worker_id selects the slot; worker_id < 0 —
    the collective merged-delta lane — uses one shared slot). Backups are
    lazily initialized to the shard's current values."""

    name = "dcasgd"

    def __init__(self, shard: torch.Tensor) -> None:
        super().__init__(shard)
        self._bak: Dict[int, torch.Tensor] = {}

    def _backup(self, worker_id: int) -> torch.Tensor:
        b = self._bak.get(worker_id)
        if b is None:
            b = self.shard.detach().clone()
            self._bak[worker_id] = b
        return b

    def update(self, delta: torch.Tensor, option: Optional[AddOption]) -> None:
        opt = option or AddOption()
        lr, lam = opt.learning_rate, opt.lambda_
        bak = self._backup(opt.worker_id)
        if self.shard.is_cuda and self.shard.dtype == torch.float32:
            _hip_ops().dcasgd_update(self.shard, bak, delta,
                                     float(lr), float(lam))
        else:
            g = delta
            self.shard.sub_(lr * (g + lam * g * g * (self.shard - bak)))
            bak.copy_(self.shard)

    def update_and_copy(self, delta, option, out) -> None:
        if self.shard.is_cuda and self.shard.dtype == torch.float32:
            opt = option or AddOption()
            bak = self._backup(opt.worker_id)
            _hip_ops().dcasgd_copy_update(self.shard, bak, delta, out,
                                          float(opt.learning_rate),
                                          float(opt.lambda_))
        else:
            self.update(delta, option)
            out.copy_(self.shard)


class DCASGDAUpdater(DCASGDUpdater):
    """DC-ASGD-a ("dcasgda", updater.cpp:52): adaptive lambda — a running
    mean-square of the gradient m = rho*m + (1-rho)*g*g scales the
    compensation coefficient to lambda / sqrt(m + eps)."""

    name = "dcasgda"
    EPS = 1e-7

    def __init__(self, shard: torch.Tensor) -> None:
        super().__init__(shard)
        self.mean_sqr = torch.zeros_like(shard)

    def update(self, delta: torch.Tensor, option: Optional[AddOption]) -> None:
        opt = option or AddOption()
        lr, lam, rho = opt.learning_rate, opt.lambda_, opt.rho
        bak = self._backup(opt.worker_id)
        if self.shard.is_cuda and self.shard.dtype == torch.float32:
            _hip_ops().dcasgda_update(self.shard, bak, self.mean_sqr, delta,
                                      float(lr), float(lam), float(rho),
                                      self.EPS)
        else:
            g = delta
            self.mean_sqr.mul_(rho).add_((1.0 - rho) * g * g)
            lam_t = lam / torch.sqrt(self.mean_sqr + self.EPS)
            self.shard.sub_(lr * (g + lam_t * g * g * (self.shard - bak)))
            bak.copy_(self.shard)

    def update_and_copy(self, delta, option, out) -> None:
        if self.shard.is_cuda and self.shard.dtype == torch.float32:
            opt = option or AddOption()
            bak = self._backup(opt.worker_id)
            _hip_ops().dcasgda_copy_update(self.shard, bak, self.mean_sqr,
                                           delta, out,
                                           float(opt.learning_rate),
                                           float(opt.lambda_),
                                           float(opt.rho), self.EPS)
        else:
            self.update(delta, option)
            out.copy_(self.shard)


_REGISTRY: Dict[str, Type[Updater]] = {
    "default": Updater,
    "sgd": SGDUpdater,
    "momentum": MomentumUpdater,
    "adagrad": AdaGradUpdater,
    "dcasgd": DCASGDUpdater,
    "dcasgda": DCASGDAUpdater,
}


def create_updater(name: str, shard: torch.Tensor) -> Updater:
    """Factory keyed by the ``updater_type`` flag (updater.cpp:46-57)."""
    try:
        cls = _REGISTRY[name]
    except KeyError:
        raise ValueError(f"unknown updater_type '{name}' "
                         f"(have {sorted(_REGISTRY)})") from None
    return cls(shard)
