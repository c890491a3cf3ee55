"""Sample readers for LogisticRegression.

Capability parity with the reference readers (reader.h:21-146, reader.cpp,
442 LoC): ``default`` libsvm-style text ("label k:v k:v ..."),
``weight``ed text ("label weight k:v ..."), and ``bsparse`` binary, all
streamed in buffered chunks with an async parse thread; the per-chunk key
set (reader ``keys()``) feeds the PS sparse pull (ps_model.cpp:290-300).
Plus a synthetic sparse generator for benchmarks (no network → no
datasets)."""

from __future__ import annotations

import struct
import threading
from queue import Queue
from typing import Iterator, List, Tuple

import torch

from .objective import Batch, DenseBatch


def parse_dense_lines(lines: List[str], weighted: bool,
                      input_size: int) -> DenseBatch:
    """Reference ``sparse=false`` text (configure.h:60-63): "label value
    value ..." — input_size-1 feature values per line plus the implicit
    bias 1 appended as the LAST column (reader.cpp:235-236). Weighted:
    "label weight value ..."."""
    rows: List[List[float]] = []
    labels: List[float] = []
    wts: List[float] = []
    for line in lines:
        parts = line.split()
        if not parts:
            continue
        labels.append(float(parts[0]))
        i = 1
        if weighted:
            wts.append(float(parts[1]))
            i = 2
        vals = [float(v) for v in parts[i:]]
        vals.append(1.0)              # bias
        rows.append(vals)
    x = torch.tensor(rows, dtype=torch.float32)
    if input_size > 0 and x.numel():
        from multiverso_amd.log import CHECK
        CHECK(x.size(1) == input_size,
              f"dense sample width {x.size(1)} (incl. bias) != "
              f"input_size {input_size}")
    return DenseBatch(x, torch.tensor(labels, dtype=torch.float32),
                      torch.tensor(wts) if weighted else None)


def parse_text_lines(lines: List[str], weighted: bool,
                     bias_key: int = -1) -> Batch:
    """default: "label k:v k:v ..." (SampleReader::ParseLine,
    reader.cpp:177-217); weight: "label weight k:v ..." with the weight
    folded into the values (WeightedSampleReader, :243-261). Every sample
    gets the bias feature key ``bias_key`` (= row_size-1) with value 1
    (:195-196, :216), when bias_key >= 0."""
    keys: List[int] = []
    vals: List[float] = []
    ptr = [0]
    labels: List[float] = []
    for line in lines:
        parts = line.split()
        if not parts:
            continue
        labels.append(float(parts[0]))
        i = 1
        w = 1.0
        if weighted:
            w = float(parts[1])
            i = 2
        for kv in parts[i:]:
            k, _, v = kv.partition(":")
            keys.append(int(k))
            vals.append((float(v) if v else 1.0) * w)
        if bias_key >= 0:
            keys.append(bias_key)
            vals.append(1.0)
        ptr.append(len(keys))
    return Batch(torch.tensor(keys, dtype=torch.int64),
                 torch.tensor(vals, dtype=torch.float32),
                 torch.tensor(ptr, dtype=torch.int64),
                 torch.tensor(labels, dtype=torch.float32))


def write_bsparse(path: str,
                  samples: List[Tuple[int, float, List[int]]]) -> None:
    """Reference bsparse binary (BSparseSampleReader::ParseSample,
    reader.cpp:391-427): per sample [size_t nnz][int32 label]
    [double weight][size_t key]*nnz; every key's value IS the weight."""
    with open(path, "wb") as f:
        for label, weight, ks in samples:
            f.write(struct.pack("<qid", len(ks), label, weight))
            for k in ks:
                f.write(struct.pack("<q", k))


def read_bsparse_batches(path: str, minibatch: int,
                         bias_key: int = -1) -> Iterator[Batch]:
    """Parses the reference layout; values = sample weight for every
    key, plus the bias key (= row_size-1, value weight) when
    bias_key >= 0 — reader.cpp:420-423."""
    keys: List[int] = []
    vals: List[float] = []
    ptr = [0]
    labels: List[float] = []
    with open(path, "rb") as f:
        while True:
            head = f.read(20)
            if len(head) < 20:
                break
            nnz, label, weight = struct.unpack("<qid", head)
            labels.append(float(label))
            for _ in range(nnz):
                (k,) = struct.unpack("<q", f.read(8))
                keys.append(k)
                vals.append(weight)
            if bias_key >= 0:
                keys.append(bias_key)
                vals.append(weight)
            ptr.append(len(keys))
            if len(labels) >= minibatch:
                yield Batch(torch.tensor(keys, dtype=torch.int64),
                            torch.tensor(vals),
                            torch.tensor(ptr, dtype=torch.int64),
                            torch.tensor(labels))
                keys, vals, ptr, labels = [], [], [0], []
    if labels:
        yield Batch(torch.tensor(keys, dtype=torch.int64),
                    torch.tensor(vals),
                    torch.tensor(ptr, dtype=torch.int64),
                    torch.tensor(labels))


class SampleReader:
    """Buffered async reader: a background thread parses the file into
    minibatch Batches (the reference's parse thread, reader.cpp)."""

    def __init__(self, path: str, minibatch: int, reader_type: str = "default",
                 buffer_batches: int = 8, input_size: int = 0,
                 sparse: bool = True) -> None:
        self.path = path
        self.minibatch = minibatch
        self.reader_type = reader_type
        self.buffer_batches = buffer_batches
        # reference `sparse` config flag (configure.h:24-25): True =
        # libsvm "label k:v ...", False = dense "label v v ..."
        self.sparse = sparse
        self.input_size = input_size
        # bias feature at row_size-1 (reader.cpp:196): requires knowing
        # the model width; 0 disables (synthetic/benchmark data)
        self.bias_key = input_size - 1 if input_size > 0 else -1

    def _produce(self, q: Queue) -> None:
        # train_file may name several files separated by ';'
        # (reference configure.h:55)
        paths = [p for p in self.path.split(";") if p]
        try:
            self._produce_inner(q, paths)
        except BaseException as e:   # surface parse errors to the consumer
            q.put(e)
        finally:
            q.put(None)

    def _produce_inner(self, q: Queue, paths) -> None:
        if True:
            if self.reader_type == "bsparse":
                for path in paths:
                    for b in read_bsparse_batches(path, self.minibatch,
                                                  self.bias_key):
                        q.put(b)
            else:
                weighted = self.reader_type == "weight"

                def flush(lines):
                    if self.sparse:
                        return parse_text_lines(lines, weighted,
                                                self.bias_key)
                    return parse_dense_lines(lines, weighted,
                                             self.input_size)
                lines: List[str] = []
                for path in paths:
                    with open(path) as f:
                        for line in f:
                            if line.strip():
                                lines.append(line)
                            if len(lines) >= self.minibatch:
                                q.put(flush(lines))
                                lines = []
                if lines:
                    q.put(flush(lines))

    def batches(self) -> Iterator[Batch]:
        q: Queue = Queue(maxsize=self.buffer_batches)
        t = threading.Thread(target=self._produce, args=(q,), daemon=True)
        t.start()
        while True:
            b = q.get()
            if b is None:
                break
            if isinstance(b, BaseException):
                raise b   # a parse error is a loud error, not truncation
            yield b


def synthetic_batches(input_size: int, n_batches: int, minibatch: int,
                      nnz: int = 32, output_size: int = 1, seed: int = 0,
                      device=None) -> Tuple[List[Batch], torch.Tensor]:
    """Synthetic sparse classification stream: a hidden sparse linear
    model labels uniform-key samples; returns (batches, true_w)."""
    device = device or torch.device("cpu")
    g = torch.Generator(device=device).manual_seed(seed)
    hidden = 4096
    true_w = torch.randn(hidden, output_size, generator=g, device=device)
    batches = []
    for i in range(n_batches):
        B = minibatch
        keys = torch.randint(0, input_size, (B * nnz,), generator=g,
                             device=device)
        vals = torch.rand(B * nnz, generator=g, device=device)
        ptr = torch.arange(0, B * nnz + 1, nnz, device=device)
        scores = torch.zeros(B, output_size, device=device)
        sid = torch.repeat_interleave(torch.arange(B, device=device), nnz)
        scores.index_add_(0, sid,
                          vals.unsqueeze(1) * true_w[keys % hidden])
        if output_size == 1:
            labels = (scores.squeeze(1) > 0).float()
        else:
            labels = scores.argmax(1).float()
        batches.append(Batch(keys, vals, ptr, labels))
    return batches, true_w
