"""Negative sampler + subsampling.

Capability parity with the reference Sampler
(Applications/WordEmbedding/src/util.cpp:116-148): 0.75-power unigram
table (kTableSize=1e8, constant.h:22), NegativeSampling via table lookup,
and frequency-based subsampling (WordSampling, util.cpp:137).

MI355X redesign: the unigram table is a device tensor resident in HBM
(400 MB at the reference size — trivial in 288 GB) so negative sampling is
a single batched gather on the GPU rather than a per-sample CPU LCG."""

from __future__ import annotations

from typing import Optional, Sequence

import torch


class Sampler:
    def __init__(self, counts: Sequence[int], table_size: int = 10_000_000,
                 power: float = 0.75,
                 device: Optional[torch.device] = None) -> None:
        device = device or torch.device("cpu")
        c = torch.as_tensor(counts, dtype=torch.float64)
        p = c.pow(power)
        p /= p.sum()
        # cumulative partition of the table (reference util.cpp:116-135)
        bounds = (p.cumsum(0) * table_size).round().long()
        table = torch.empty(table_size, dtype=torch.int64)
        start = 0
        for wid, end in enumerate(bounds.tolist()):
            end = min(end, table_size)
            if end > start:
                table[start:end] = wid
            start = end
        if start < table_size:
            table[start:] = len(counts) - 1
        self.table = table.to(device)
        self.counts = c.to(device)
        self.total = float(c.sum())

    def negative_sampling(self, shape, generator=None) -> torch.Tensor:
        idx = torch.randint(0, self.table.numel(), shape,
                            device=self.table.device, generator=generator)
        return self.table[idx]

    def keep_mask(self, words: torch.Tensor, sample: float,
                  generator=None) -> torch.Tensor:
        """Subsampling keep-probability mask (util.cpp:137-143):
        p_keep = (sqrt(f/(sample)) + 1) * sample/f with f the corpus
        frequency of the word."""
        if sample <= 0:
            return torch.ones_like(words, dtype=torch.bool)
        f = self.counts[words] / self.total
        p = (torch.sqrt(f / sample) + 1) * sample / f
        r = torch.rand(words.shape, device=words.device, generator=generator)
        return r < p.clamp(max=1.0)
