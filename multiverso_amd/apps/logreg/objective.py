"""Objectives + regularizers for LogisticRegression.

Capability parity with the reference objective family
(Applications/LogisticRegression/src/objective/objective.cpp): sigmoid
(:151-188), softmax (:193-230), FTRL (:250-345 — z/n entry state, w
reconstruction at predict, delta_z/delta_n gradients), factory by
``objective_type``; L1/L2 regularizers (regular/l1_regular.h,
l2_regular.h) added into the gradient (AddRegularization,
objective.cpp:63-100).

All math is batched torch over a sparse minibatch
(keys[nnz], vals[nnz], ptr[B+1], labels[B]) — the reference's per-sample
loops (K13) become segment reductions that run on the GPU."""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from multiverso_amd import ops


class DenseBatch:
    """Dense minibatch (reference ``sparse=false`` data,
    configure.h:60-63: "label value value ..."): ``x`` is [B, d]
    float32 with the bias column (value 1) LAST — the reference appends
    the bias value to every sample (reader.cpp:235-236)."""

    __slots__ = ("x", "labels", "weights")

    def __init__(self, x: torch.Tensor, labels: torch.Tensor,
                 weights: Optional[torch.Tensor] = None) -> None:
        self.x = x
        self.labels = labels
        self.weights = weights

    @property
    def size(self) -> int:
        return self.x.size(0)

    def to(self, device) -> "DenseBatch":
        return DenseBatch(self.x.to(device), self.labels.to(device),
                          None if self.weights is None
                          else self.weights.to(device))


class Batch:
    """Sparse minibatch: sample i has keys[ptr[i]:ptr[i+1]] etc."""

    __slots__ = ("keys", "vals", "ptr", "labels", "weights")

    def __init__(self, keys: torch.Tensor, vals: torch.Tensor,
                 ptr: torch.Tensor, labels: torch.Tensor,
                 weights: Optional[torch.Tensor] = None) -> None:
        self.keys = keys
        self.vals = vals
        self.ptr = ptr
        self.labels = labels
        self.weights = weights

    @property
    def size(self) -> int:
        return self.ptr.numel() - 1

    def to(self, device) -> "Batch":
        return Batch(self.keys.to(device), self.vals.to(device),
                     self.ptr.to(device), self.labels.to(device),
                     None if self.weights is None
                     else self.weights.to(device))

    def sample_ids(self) -> torch.Tensor:
        lens = self.ptr[1:] - self.ptr[:-1]
        return torch.repeat_interleave(
            torch.arange(self.size, device=self.keys.device), lens)


def _scores(batch, w_rows: torch.Tensor) -> torch.Tensor:
    """scores[B, O] = sum_j x_j * W[key_j, :] per sample (K13).

    Dense batches: one GEMM, X[B,d] @ W[d,O] — routed through
    torch.matmul = rocBLAS on ROCm, i.e. the MFMA matrix cores (the
    library is the right owner of compute-bound plain GEMMs; see
    docs/ENGINEERING_NOTES.md "MFMA decision"). ``w_rows`` is then the
    FULL weight matrix.

    Sparse batches are CSR (ptr segments a sample's features
    contiguously), so the per-sample sum is a segmented reduction:
    cumsum + boundary diff — no atomics (measured 3x the throughput of
    the scatter-add form on gfx950; fp64 accumulator keeps the diff
    exact for fp32 data)."""
    if isinstance(batch, DenseBatch):
        return batch.x @ w_rows
    contrib = (batch.vals.unsqueeze(1) * w_rows).to(torch.float64)
    cs = torch.empty(contrib.size(0) + 1, contrib.size(1),
                     device=contrib.device, dtype=torch.float64)
    cs[0].zero_()
    torch.cumsum(contrib, 0, out=cs[1:])
    ptr = batch.ptr.long()
    return (cs[ptr[1:]] - cs[ptr[:-1]]).to(w_rows.dtype)


def _one_hot(labels: torch.Tensor, O: int) -> torch.Tensor:
    if O == 1:
        return labels.float().unsqueeze(1)
    return torch.nn.functional.one_hot(labels.long(), O).float()


class Regularizer:
    def delta(self, w_rows: torch.Tensor) -> torch.Tensor:
        return torch.zeros_like(w_rows)


class L1Regular(Regularizer):
    def __init__(self, coef: float) -> None:
        self.coef = coef

    def delta(self, w_rows: torch.Tensor) -> torch.Tensor:
        return self.coef * torch.sign(w_rows)


class L2Regular(Regularizer):
    def __init__(self, coef: float) -> None:
        self.coef = coef

    def delta(self, w_rows: torch.Tensor) -> torch.Tensor:
        return self.coef * w_rows


def create_regularizer(cfg) -> Regularizer:
    if cfg.regular_type == "l1":
        return L1Regular(cfg.regular_coef)
    if cfg.regular_type == "l2":
        return L2Regular(cfg.regular_coef)
    return Regularizer()


class Objective:
    """Gradient returns (grad_rows[nnz, O], train_loss). grad_rows is the
    per-(key occurrence, output) gradient x*loss; the caller aggregates
    into its weight delta."""

    def __init__(self, cfg) -> None:
        self.cfg = cfg
        self.output_size = cfg.output_size
        self.regular = create_regularizer(cfg)

    def predict(self, batch: Batch, w_rows: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def gradient(self, batch, w_rows: torch.Tensor
                 ) -> Tuple[torch.Tensor, float]:
        p = self.predict(batch, w_rows)
        diff = p - _one_hot(batch.labels, self.output_size)
        if batch.weights is not None:
            diff = diff * batch.weights.unsqueeze(1)
        if isinstance(batch, DenseBatch):
            # dense: grad[d, O] = X^T @ diff — the second GEMM of the
            # pair (also rocBLAS/MFMA); regularizer applies to the full
            # weight matrix (reference AddRegularization over the whole
            # dense model)
            grad = batch.x.t() @ diff + self.regular.delta(w_rows)
            return grad, self.loss(batch, p)
        grad = batch.vals.unsqueeze(1) * diff[batch.sample_ids()]
        grad = grad + self.regular.delta(w_rows)
        loss = self.loss(batch, p)
        return grad, loss

    def loss(self, batch: Batch, p: torch.Tensor) -> float:
        y = _one_hot(batch.labels, self.output_size)
        eps = 1e-12
        ll = -(y * torch.log(p + eps)
               + (1 - y) * torch.log(1 - p + eps)).sum(1)
        return float(ll.mean())

    def correct(self, batch: Batch, p: torch.Tensor) -> int:
        if self.output_size == 1:
            pred = (p.squeeze(1) > 0.5).long()
            return int((pred == batch.labels.long()).sum())
        return int((p.argmax(1) == batch.labels.long()).sum())


class SigmoidObjective(Objective):
    """objective.cpp:151-188."""

    def predict(self, batch: Batch, w_rows: torch.Tensor) -> torch.Tensor:
        return torch.sigmoid(_scores(batch, w_rows))


class SoftmaxObjective(Objective):
    """objective.cpp:193-230."""

    def predict(self, batch: Batch, w_rows: torch.Tensor) -> torch.Tensor:
        return torch.softmax(_scores(batch, w_rows), dim=1)

    def loss(self, batch: Batch, p: torch.Tensor) -> float:
        eps = 1e-12
        y = batch.labels.long()
        return float(-torch.log(p[torch.arange(p.size(0),
                                               device=p.device), y]
                                + eps).mean())


class FTRLObjective(Objective):
    """objective.cpp:250-345. Model rows carry (z, n) interleaved:
    w_rows[:, 0:O] = z, w_rows[:, O:2O] = n. gradient() returns
    (delta_z | delta_n) rows to be SUBTRACTED from the entry state by the
    server's ftrl updater (updater.cpp:79-101: z -= dz; n -= dn)."""

    def __init__(self, cfg) -> None:
        super().__init__(cfg)
        self.alpha_inv = 1.0 / cfg.alpha
        self.beta = cfg.beta
        self.l1 = cfg.lambda1
        self.l2 = cfg.lambda2

    def reconstruct_w(self, zn: torch.Tensor) -> torch.Tensor:
        O = self.output_size
        z, n = zn[:, :O], zn[:, O:]
        sqrtn = torch.sqrt(torch.clamp(n, min=0.0))
        w = (torch.sign(z) * self.l1 - z) / (
            (self.beta + sqrtn) * self.alpha_inv + self.l2)
        return torch.where(z.abs() > self.l1, w, torch.zeros_like(w))

    def predict(self, batch: Batch, zn_rows: torch.Tensor) -> torch.Tensor:
        return torch.sigmoid(_scores(batch, self.reconstruct_w(zn_rows)))

    def gradient(self, batch, zn_rows: torch.Tensor
                 ) -> Tuple[torch.Tensor, float]:
        if isinstance(batch, DenseBatch):
            # FTRL's per-occurrence z/n state math is sparse by
            # construction (the reference pairs FTRL with its sparse
            # tables, ftrl_sparse_table.h); refuse dense loudly.
            from multiverso_amd.log import CHECK
            CHECK(False, "FTRL objective requires sparse data "
                         "(objective_type=ftrl with sparse=false)")
        O = self.output_size
        w = self.reconstruct_w(zn_rows)
        p = torch.sigmoid(_scores(batch, w))
        diff = p - _one_hot(batch.labels, O)
        g = batch.vals.unsqueeze(1) * diff[batch.sample_ids()]  # delta_g
        g2 = g * g
        n = torch.clamp(zn_rows[:, O:], min=0.0)
        sqrtn = torch.sqrt(n)
        delta_z = (self.alpha_inv * (torch.sqrt(n + g2) - sqrtn)) * w - g
        delta_n = -g2
        return torch.cat([delta_z, delta_n], dim=1), self.loss(batch, p)


def create_objective(cfg) -> Objective:
    t = cfg.objective_type
    if t == "sigmoid":
        return SigmoidObjective(cfg)
    if t == "softmax":
        return SoftmaxObjective(cfg)
    if t == "ftrl":
        return FTRLObjective(cfg)
    raise ValueError(f"unknown objective_type '{t}'")
