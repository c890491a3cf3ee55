"""LogReg driver: train/test/save/load.

Capability parity with the reference LogReg<float>
(Applications/LogisticRegression/src/logreg.cpp): config-file
construction, epoch training loop with async reader and periodic loss /
time-per-sample logging (:40-90), Test with prediction output (:119-173),
SaveModel/LoadModel (:92-117)."""

from __future__ import annotations

import time
from typing import Iterator, List, Optional


import multiverso_amd as mv

from .config import LogRegConfig
from .model import create_model
from .objective import Batch
from .reader import SampleReader


class LogReg:
    def __init__(self, config, device=None) -> None:
        if isinstance(config, str):
            config = LogRegConfig.from_file(config)
        self.cfg = config
        if config.use_ps and not mv.Zoo.get().started:
            mv.init()
        self.model = create_model(config, device)
        if config.init_model_file:
            self.model.load(config.init_model_file)

    # ---- training (logreg.cpp:40-90) ----
    def train(self, batches: Optional[Iterator[Batch]] = None) -> float:
        cfg = self.cfg
        last_loss = 0.0
        for epoch in range(cfg.train_epoch):
            it = batches if batches is not None else SampleReader(
                cfg.train_file, cfg.minibatch_size, cfg.reader_type,
                input_size=cfg.input_size, sparse=cfg.sparse).batches()
            t0 = time.perf_counter()
            nsamples = 0
            losses: List[float] = []

            def show(b_size: int) -> None:
                if (cfg.show_time_per_sample
                        and nsamples % cfg.show_time_per_sample < b_size):
                    dt = time.perf_counter() - t0
                    mv.log.info(f"[logreg] epoch {epoch} samples {nsamples} "
                                f"loss {sum(losses)/max(len(losses),1):.5f} "
                                f"{dt/max(nsamples,1)*1e6:.2f} us/sample")

            if not cfg.use_ps:
                for b in it:
                    nsamples += b.size
                    losses.append(self.model.update(b))
                    show(b.size)
            else:
                def chunks():
                    chunk: List[Batch] = []
                    for b in it:
                        chunk.append(b)
                        if len(chunk) >= cfg.sync_frequency:
                            yield chunk
                            chunk = []
                    if chunk:
                        yield chunk

                ci = chunks()
                while True:
                    c = next(ci, None)
                    if mv.size() > 1:
                        # train_chunk is collective and ranks may hold
                        # different sample counts (per-rank train files,
                        # configure.h:55): exhausted ranks keep joining
                        # with empty chunks until every rank is done.
                        import torch
                        have = torch.tensor([0.0 if c is None else 1.0])
                        mv.aggregate(have)
                        if float(have[0]) == 0.0:
                            break
                        c = c or []
                    elif c is None:
                        break
                    nsamples += sum(b.size for b in c)
                    losses.append(self.model.train_chunk(c))
                    show(sum(b.size for b in c) or 1)
            last_loss = sum(losses) / max(len(losses), 1)
            mv.log.info(f"[logreg] epoch {epoch} done: avg loss "
                        f"{last_loss:.5f}")
            if batches is not None and cfg.train_epoch > 1:
                batches = None  # generator is exhausted; re-read from file
        return last_loss

    # ---- evaluation (logreg.cpp:119-173) ----
    def test(self, batches: Optional[Iterator[Batch]] = None,
             output_file: Optional[str] = None):
        cfg = self.cfg
        it = batches if batches is not None else SampleReader(
            cfg.test_file, cfg.minibatch_size, cfg.reader_type,
            input_size=cfg.input_size, sparse=cfg.sparse).batches()
        correct = total = 0
        loss_sum = 0.0
        nb = 0
        out = open(output_file, "w") if output_file else None
        collective = cfg.use_ps and mv.size() > 1
        itr = iter(it)
        while True:
            b = next(itr, None)
            if collective:
                # predict pulls rows collectively; ranks may hold
                # different test-set sizes -- empty-batch participation
                # until all ranks are done (same protocol as train)
                import torch
                have = torch.tensor([0.0 if b is None else 1.0])
                mv.aggregate(have)
                if float(have[0]) == 0.0:
                    break
                if b is None:
                    if cfg.sparse:
                        e = torch.empty(0, dtype=torch.int64)
                        b = Batch(e, torch.empty(0),
                                  torch.zeros(1, dtype=torch.int64),
                                  torch.empty(0))
                    else:
                        from .objective import DenseBatch
                        b = DenseBatch(
                            torch.empty(0, cfg.input_size),
                            torch.empty(0))
            elif b is None:
                break
            p = self.model.predict(b)
            if b.size == 0:
                continue
            correct += self.model.objective.correct(b.to(p.device), p)
            loss_sum += self.model.objective.loss(b.to(p.device), p)
            nb += 1
            total += b.size
            if out is not None:
                for row in p.cpu().tolist():
                    out.write(" ".join(f"{x:.6f}" for x in row) + "\n")
        if out is not None:
            out.close()
        acc = correct / max(total, 1)
        mv.log.info(f"[logreg] test: {correct}/{total} correct "
                    f"({acc:.4f}), avg loss {loss_sum / max(nb, 1):.5f}")
        return acc, loss_sum / max(nb, 1)

    def save_model(self, path: Optional[str] = None) -> None:
        self.model.store(path or self.cfg.output_model_file)

    def load_model(self, path: Optional[str] = None) -> None:
        self.model.load(path or self.cfg.init_model_file)
