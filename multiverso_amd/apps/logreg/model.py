"""LogReg models: local and parameter-server backed.

Capability parity with the reference Model/PSModel
(Applications/LogisticRegression/src/model/model.cpp:62-222,
ps_model.cpp): minibatch delta accumulation + updater apply; the PS
variant holds weights in a table (Array/Sparse/FTRL by config,
ps_model.cpp:26-41), sets multiverso ``updater_type=sgd`` (:24), syncs
every ``sync_frequency`` minibatches (:172-182), optionally pipelines the
pull with a double buffer (:236-271), and pre-pulls the whole model
before store (:157-169).

MI355X mapping: weights live in a row-sharded MatrixTable
(rows = input_size features, cols = output_size; FTRL cols = 2*output for
(z|n)); the per-chunk pull is get_rows (all-to-all over xGMI, K6 gather),
the push is add_rows (K5/K15 scatter-update). Dense mode uses whole-table
collectives. Worker-side lr decay matches SGDUpdater
(updater.cpp:53-71): lr = max(1e-3, lr0 - count/(coef*minibatch))."""

from __future__ import annotations

from typing import List

import numpy as np
import torch

import multiverso_amd as mv
from multiverso_amd import ops

from .objective import Batch, DenseBatch, create_objective


class WorkerSGDSchedule:
    def __init__(self, cfg) -> None:
        self.lr0 = cfg.learning_rate
        self.coef = cfg.learning_rate_coef
        self.minibatch = cfg.minibatch_size
        self.count = 0

    def next_lr(self) -> float:
        lr = max(1e-3, self.lr0 - self.count / (self.coef * self.minibatch))
        self.count += 1
        return lr


class LocalModel:
    """use_ps=false: full weight tensor on this device
    (model/model.cpp:62-135)."""

    def __init__(self, cfg, device=None) -> None:
        self.cfg = cfg
        self.device = device or torch.device("cpu")
        self.objective = create_objective(cfg)
        cols = cfg.output_size * (2 if cfg.objective_type == "ftrl" else 1)
        self.weight = torch.zeros(cfg.input_size, cols, device=self.device)
        self.sched = WorkerSGDSchedule(cfg)

    def update(self, batch) -> float:
        batch = batch.to(self.device)
        if isinstance(batch, DenseBatch):
            # dense data (reference sparse=false): whole-matrix GEMM
            # gradient on the matrix cores via rocBLAS
            grad, loss = self.objective.gradient(batch, self.weight)
            self.weight -= self.sched.next_lr() * grad
            return loss
        w_rows = self.weight[batch.keys]
        grad, loss = self.objective.gradient(batch, w_rows)
        if self.cfg.objective_type == "ftrl":
            # server/local ftrl update is unscaled: state -= delta
            ops.scatter_add_rows(self.weight, batch.keys, grad, -1.0)
        else:
            lr = self.sched.next_lr()
            ops.scatter_add_rows(self.weight, batch.keys, grad, -lr)
        return loss

    def predict(self, batch) -> torch.Tensor:
        batch = batch.to(self.device)
        if isinstance(batch, DenseBatch):
            return self.objective.predict(batch, self.weight)
        return self.objective.predict(batch, self.weight[batch.keys])

    # ---- model file io (model.cpp:146-204 / sparse_table.h:258-285) ----
    def store(self, path: str) -> None:
        w = self.weight.cpu().numpy()
        if self.cfg.sparse:
            nz = np.nonzero(w.any(axis=1))[0].astype(np.int64)
            with open(path, "wb") as f:
                f.write(np.int64(nz.size).tobytes())
                f.write(nz.tobytes())
                f.write(w[nz].astype(np.float32).tobytes())
        else:
            w.astype(np.float32).tofile(path)

    def load(self, path: str) -> None:
        cols = self.weight.shape[1]
        if self.cfg.sparse:
            with open(path, "rb") as f:
                cnt = int(np.frombuffer(f.read(8), dtype=np.int64)[0])
                keys = np.frombuffer(f.read(8 * cnt), dtype=np.int64)
                vals = np.frombuffer(f.read(4 * cnt * cols),
                                     dtype=np.float32).reshape(cnt, cols)
            self.weight.zero_()
            self.weight[torch.from_numpy(keys.copy())] = \
                torch.from_numpy(vals.copy()).to(self.device)
        else:
            w = np.fromfile(path, dtype=np.float32).reshape(
                self.weight.shape)
            self.weight.copy_(torch.from_numpy(w).to(self.device))


class PSModel:
    """use_ps=true: weights in a row-sharded table; chunked train with
    pull → local minibatch steps → push (ps_model.cpp)."""

    def __init__(self, cfg, device=None) -> None:
        self.cfg = cfg
        self.device = device or mv.Zoo.get().device
        self.objective = create_objective(cfg)
        self.is_ftrl = cfg.objective_type == "ftrl"
        cols = cfg.output_size * (2 if self.is_ftrl else 1)
        updater = "sgd" if (self.is_ftrl or cfg.updater_type
                            in ("sgd", "default", "ftrl")) else cfg.updater_type
        self.table = mv.MatrixTable(cfg.input_size, cols,
                                    updater_type=updater)
        self.sched = WorkerSGDSchedule(cfg)
        self.cols = cols

    def train_chunk(self, batches: List[Batch]) -> float:
        """One sync group: pull union keys, run the minibatches against
        the pulled snapshot (within-chunk updates visible locally), push
        the summed delta (ps_model.cpp:172-203)."""
        if not self.cfg.sparse:
            return self._train_chunk_dense(batches)
        batches = [b.to(self.device) for b in batches]
        if batches:
            cat_keys = torch.cat([b.keys for b in batches])
            if self.cfg.input_size <= (1 << 31) - 1:
                # keys fit int32: the dedup sort runs at half the bytes.
                # (An explicit torch.sort + manual inverse/masked-select
                # was measured 10-15% SLOWER whole-app than unique —
                # the extra elementwise/scatter kernels outweigh any
                # sort-path difference; keep unique.)
                union, inv = torch.unique(cat_keys.to(torch.int32),
                                          return_inverse=True)
                union = union.to(torch.int64)
            else:
                union, inv = torch.unique(cat_keys, return_inverse=True)
            # the inverse IS each occurrence's local row in the pulled
            # buffer — split per minibatch and the per-batch searchsorted
            # (4x ~21 us/chunk on MI355X) disappears
            sizes = [b.keys.numel() for b in batches]
            lidxs = list(torch.split(inv, sizes))
        else:
            union = torch.empty(0, dtype=torch.int64, device=self.device)
            lidxs = []
        pulled = self.table.get_rows(union)
        local = pulled.clone()
        total_loss = 0.0
        adagrad = self.table.updater_type == "adagrad"
        fused = self._fused_kind(local)
        if fused:
            total_loss = self._train_chunk_fused(batches, lidxs, local,
                                                 fused)
        else:
            for b, lidx in zip(batches, lidxs):
                w_rows = local[lidx]
                grad, loss = self.objective.gradient(b, w_rows)
                total_loss += loss
                if self.is_ftrl:
                    ops.scatter_add_rows(local, lidx, grad, -1.0)
                elif adagrad:
                    # server-side adagrad consumes lr-scaled deltas;
                    # locally approximate with plain sgd steps for
                    # within-chunk vis.
                    lr = self.sched.next_lr()
                    ops.scatter_add_rows(local, lidx, grad, -lr)
                else:
                    lr = self.sched.next_lr()
                    ops.scatter_add_rows(local, lidx, grad, -lr)
        if adagrad:
            delta = pulled - local   # = sum(lr*grad); server g=delta/lr
            opt = mv.AddOption(learning_rate=1.0, rho=self.cfg.learning_rate)
            self.table.add_rows(union, delta, option=opt,
                                assume_unique=True)
        else:
            # server updater 'sgd': w -= delta; push accumulated movement
            self.table.add_rows(union, pulled - local, assume_unique=True)
        return total_loss / max(len(batches), 1)

    def _train_chunk_dense(self, batches) -> float:
        """Dense chunk (reference ``sparse=false`` + use_ps — its own
        mnist.config deployment): the model is pulled WHOLE (all-gather
        of HBM shards), each minibatch is two GEMMs (scores = X@W,
        grad = X^T@diff) on the matrix cores via rocBLAS, and the
        summed movement pushes back as ONE whole-table add
        (reduce-scatter + updater kernel) — the bandwidth plane end to
        end. Empty chunks still participate collectively."""
        from multiverso_amd.log import CHECK
        batches = [b.to(self.device) for b in batches]
        pulled = self.table.get()
        local = pulled.clone()
        for b in batches:
            CHECK(isinstance(b, DenseBatch),
                  "cfg.sparse=false requires dense batches "
                  "(\"label value value ...\" data)")
        if local.is_cuda and self._dense_fused_ok():
            total_loss = self._dense_minibatches_fused(batches, local)
        else:
            total_loss = 0.0
            for b in batches:
                grad, loss = self.objective.gradient(b, local)
                total_loss += loss
                local -= self.sched.next_lr() * grad
        if self.table.updater_type == "adagrad":
            opt = mv.AddOption(learning_rate=1.0,
                               rho=self.cfg.learning_rate)
            self.table.add(pulled - local, option=opt).wait()
        else:
            self.table.add(pulled - local).wait()
        return total_loss / max(len(batches), 1)

    def _dense_fused_ok(self) -> bool:
        """GPU dense minibatch fusion applies to the sigmoid/softmax
        objectives (K <= 64).  FTRL dense is refused upstream."""
        from .objective import SigmoidObjective, SoftmaxObjective
        return (type(self.objective) in (SigmoidObjective,
                                         SoftmaxObjective)
                and self.cols <= 64)

    def _dense_minibatches_fused(self, batches, local) -> float:
        """Dense minibatch loop as 4 GPU ops each (GEMM, fused
        post-kernel, GEMM, update) instead of ~15: scores = X@W stays on
        rocBLAS/MFMA, then ``lr_dense_post`` turns the logits buffer
        into the (p - onehot(y))*wt diff IN PLACE and atomically
        accumulates the per-batch mean loss (one host sync per chunk),
        then grad = X^T@diff (rocBLAS again) and the SGD/regularizer
        update runs as in-place torch ops.  Numerics match
        objective.gradient + the learning-rate step
        (tests/test_gpu_kernels.py::test_lr_dense_post_*)."""
        from multiverso_amd import ops as _ops
        from .objective import L1Regular, L2Regular
        hip = _ops.module(required=True)
        reg = self.objective.regular
        loss_acc = torch.zeros((), device=self.device)
        logits = None   # reused across equal-sized minibatches
        for b in batches:
            B = b.size
            if B == 0:
                self.sched.next_lr()
                continue
            wts = None if b.weights is None else b.weights.float()
            labels = (b.labels if b.labels.dtype == torch.float32
                      else b.labels.float())
            if logits is None or logits.shape[0] != B:
                logits = torch.empty(B, self.cols, device=self.device)
            if not hip.lr_dense_fwd(b.x, local, labels, wts,
                                    logits, loss_acc, 1.0 / B):
                # d*K over the LDS budget: rocBLAS GEMM + fused post
                torch.matmul(b.x, local, out=logits)
                hip.lr_dense_post(logits, labels, wts, loss_acc,
                                  1.0 / B)
            lr = self.sched.next_lr()
            # backward stays on rocBLAS: a hand-rolled X^T@diff was
            # measured 4-17x SLOWER (ENGINEERING_NOTES "dense backward")
            if isinstance(reg, L2Regular):
                local.mul_(1.0 - lr * reg.coef)
            elif isinstance(reg, L1Regular):
                local.add_(torch.sign(local), alpha=-lr * reg.coef)
            local.addmm_(b.x.t(), logits, alpha=-lr)
        return float(loss_acc)

    def _fused_kind(self, local: torch.Tensor) -> str:
        """Which fused K13/K14 minibatch kernel pair serves this
        objective on GPU ('' = torch path). All three reference
        objectives now have fused forms: sigmoid (1 col), softmax
        (2..64 classes), ftrl (z|n state, <=32 outputs)."""
        from .objective import (FTRLObjective, SigmoidObjective,
                                SoftmaxObjective)
        if not local.is_cuda:
            return ""
        if self.cols == 1 and type(self.objective) is SigmoidObjective:
            return "sigmoid"
        if (type(self.objective) is SoftmaxObjective
                and 2 <= self.cols <= 64):
            return "softmax"
        if (type(self.objective) is FTRLObjective
                and self.cfg.output_size <= 32):
            return "ftrl"
        return ""

    def _train_chunk_fused(self, batches, lidxs, local, kind) -> float:
        """Two HIP kernels per minibatch (fwd + scatter) instead of ~20
        torch ops; numerics match objective.gradient + scatter
        (tests/test_gpu_kernels.py). Returns summed per-batch mean
        losses (one host sync per chunk instead of per minibatch)."""
        from multiverso_amd import ops as _ops
        hip = _ops.module(required=True)
        from .objective import L1Regular, L2Regular
        reg = self.objective.regular
        reg_type, reg_coef = 0, 0.0
        if isinstance(reg, L1Regular):
            reg_type, reg_coef = 1, reg.coef
        elif isinstance(reg, L2Regular):
            reg_type, reg_coef = 2, reg.coef
        K = self.cfg.output_size
        wflat = local.view(-1)
        loss_acc = torch.zeros((), device=self.device)
        for b, lidx in zip(batches, lidxs):
            ptr32 = b.ptr.to(torch.int32)
            B = b.size
            lossb = torch.empty(B, device=self.device)
            wts = None if b.weights is None else b.weights.float()
            labels = b.labels.float()
            if kind == "sigmoid":
                err = torch.empty(B, device=self.device)
                hip.lr_sigmoid_forward(wflat, lidx, b.vals, ptr32, labels,
                                       wts, err, lossb)
                lr = self.sched.next_lr()
                hip.lr_sigmoid_scatter(wflat, lidx, b.vals, ptr32, err, lr,
                                       reg_type, reg_coef)
            elif kind == "softmax":
                err = torch.empty(B * K, device=self.device)
                hip.lr_softmax_forward(wflat, lidx, b.vals, ptr32, labels,
                                       wts, err, lossb, K)
                lr = self.sched.next_lr()
                hip.lr_softmax_scatter(wflat, lidx, b.vals, ptr32, err, lr,
                                       reg_type, reg_coef, K)
            else:  # ftrl: state update is lr-free (updater.cpp:79-101)
                o = self.objective
                err = torch.empty(B * K, device=self.device)
                hip.lr_ftrl_forward(wflat, lidx, b.vals, ptr32, labels, wts,
                                    err, lossb, 1.0 / o.alpha_inv, o.beta,
                                    o.l1, o.l2, K)
                hip.lr_ftrl_scatter(wflat, lidx, b.vals, ptr32, err,
                                    1.0 / o.alpha_inv, o.beta, o.l1, o.l2,
                                    K)
            loss_acc += lossb.mean()
        return float(loss_acc)

    def predict(self, batch) -> torch.Tensor:
        batch = batch.to(self.device)
        if isinstance(batch, DenseBatch):
            # whole-model pull (collective — dense ranks' empty
            # participation batches are DenseBatches too)
            return self.objective.predict(batch, self.table.get())
        w_rows = self.table.get_rows(batch.keys)
        return self.objective.predict(batch, w_rows)

    # ---- store/load via the whole-model pre-pull (ps_model.cpp:157-169) --
    def store(self, path: str) -> None:
        w = self.table.get()
        if mv.rank() == 0:
            w.cpu().numpy().astype(np.float32).tofile(path)
        mv.barrier()

    def load(self, path: str) -> None:
        w = np.fromfile(path, dtype=np.float32).reshape(
            self.cfg.input_size, self.cols)
        full = torch.from_numpy(w).to(self.device)
        sl = full[self.table.row_offset:
                  self.table.row_offset + self.table.local_rows]
        self.table.shard.copy_(sl)
        mv.barrier()


def create_model(cfg, device=None):
    """model.cpp:216-222 factory."""
    if cfg.use_ps:
        return PSModel(cfg, device)
    return LocalModel(cfg, device)
