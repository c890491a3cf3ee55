"""Distributed word2vec on the parameter server.

Capability parity with the reference WordEmbedding application
(Applications/WordEmbedding/src): skip-gram and CBOW,
negative sampling and hierarchical softmax, optional per-element AdaGrad,
subsampling, linear lr decay driven by a globally-synced word count
(KV table), block-pipelined training, and word2vec-format embedding save
(distributed_wordembedding.cpp:263-306).

MI355X-native redesign of the training loop:
- The per-sentence scalar loops (wordembedding.cpp:17-166) become ONE
  fused HIP kernel per data block (ops/csrc/kernels.hip k_w2v): one wave
  per training group, dot-products wave-reduced, row updates atomic.
- Pair extraction, reduced-window masking, negative sampling and the
  global→block-local row mapping run as batched torch ops on the GPU
  (PrepareData, wordembedding.cpp:169-213, without the CPU loops).
- Parameter pull/push per block uses the row-keyed table ops
  (get_rows/add_rows = all-to-all over xGMI; the reference's
  RequestParameter/AddDeltaParameter, communicator.cpp:117-249), with
  delta = (trained - pulled)/num_workers (communicator.cpp:167).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch

import multiverso_amd as mv

from .huffman import HuffmanEncoder
from .sampler import Sampler


@dataclass
class WordEmbeddingOption:
    """CLI-parity options (reference util.cpp:6-57)."""
    embedding_size: int = 200
    window: int = 5
    negative_num: int = 5
    hs: bool = False
    cbow: bool = False
    min_count: int = 5
    sample: float = 0.0
    init_learning_rate: float = 0.025
    epoch: int = 1
    use_adagrad: bool = False
    total_words: int = 0
    data_block_size: int = 100_000  # words per block (ref default 1e6)
    unigram_table_size: int = 10_000_000
    seed: int = 1
    # Within a kernel launch, groups train hogwild-concurrently (row reads
    # are pre-update, like the reference's unsynchronized OpenMP trainers,
    # which give no ordering guarantee within a block either). Setting this
    # below a block's group count chunks the block into sequential launches
    # so later occurrences of a pair see earlier updates — bounded
    # staleness, STRICTER than reference semantics; useful for tiny
    # duplicate-heavy corpora (see tests).
    max_groups_per_launch: int = 1 << 23
    # Row updates race hogwild-style by default (the reference's own
    # unsynchronized OpenMP semantics; 11x faster than atomics on
    # MI355X). Set True for exact atomic accumulation.
    atomic_updates: bool = False


class WordEmbedding:
    def __init__(self, option: WordEmbeddingOption, counts: List[int],
                 device: Optional[torch.device] = None) -> None:
        self.opt = option
        self.vocab_size = len(counts)
        self.device = device or mv.Zoo.get().device
        dim = option.embedding_size
        # Tables (reference constant.h:16-20 / communicator.cpp:17-32):
        # input embeddings random-init U[-0.5/dim, 0.5/dim), output zeros.
        bound = 0.5 / dim
        self.input_table = mv.MatrixTable(self.vocab_size, dim,
                                          random_init=(-bound, bound))
        self.output_table = mv.MatrixTable(self.vocab_size, dim)
        self.input_gsq_table = self.output_gsq_table = None
        if option.use_adagrad:
            self.input_gsq_table = mv.MatrixTable(self.vocab_size, dim)
            self.output_gsq_table = mv.MatrixTable(self.vocab_size, dim)
        self.word_count_table = mv.KVTable()
        self.sampler = Sampler(counts, option.unigram_table_size,
                               device=self.device)
        self.huffman: Optional[HuffmanEncoder] = None
        self._hs_point = self._hs_label = self._hs_off = None
        if option.hs:
            self.huffman = HuffmanEncoder()
            self.huffman.build_from_term_frequency(counts)
            self._build_hs_tensors()
        self.learning_rate = option.init_learning_rate
        self._block_pool: Optional[torch.Tensor] = None
        self._seed_state = (option.seed * 0x2545F4914F6CDD1D) & 0xFFFFFFFFFFFFFFFF
        self.word_count_local = 0      # words since last global sync
        self.word_count_actual = 0     # global processed words
        self.gen = torch.Generator(device=self.device)
        self.gen.manual_seed(option.seed + 7919 * mv.rank())

    def _build_hs_tensors(self) -> None:
        pts, labels, off = [], [], [0]
        for info in self.huffman.labels:
            pts.extend(info.point)
            labels.extend(1 - c for c in info.code)  # error = (1-code) - f
            off.append(len(pts))
        self._hs_point = torch.tensor(pts, dtype=torch.int64,
                                      device=self.device)
        self._hs_label = torch.tensor(labels, dtype=torch.float32,
                                      device=self.device)
        self._hs_off = torch.tensor(off, dtype=torch.int64,
                                    device=self.device)

    # ------------------------------------------------------------------
    # Group construction (PrepareData equivalent, fully batched on GPU)
    # ------------------------------------------------------------------
    def build_pairs(self, words: torch.Tensor, sent_ids: torch.Tensor):
        """Returns (in_idx, in_off, centers) with GLOBAL word ids: one
        group per training sample, inputs per group via in_off ragged
        offsets, `centers` the positive output of each group."""
        opt = self.opt
        if opt.cbow:
            return self._build_cbow_pairs(words, sent_ids)
        device = self.device
        n = words.numel()
        W = min(opt.window, n - 1)
        empty = (torch.empty(0, dtype=torch.int64, device=device),
                 torch.zeros(1, dtype=torch.int32, device=device),
                 torch.empty(0, dtype=torch.int64, device=device))
        if W < 1:
            return empty
        # all window offsets in one batched pass: pair (i, i+o) for
        # o = 1..W, valid when in-range + same sentence + reduced-window
        # keep (offset o survives with prob (window-o+1)/window)
        offs = torch.arange(1, W + 1, device=device)            # (W,)
        left = torch.arange(n, device=device).unsqueeze(1)      # (n,1)
        right = left + offs                                     # (n,W)
        inb = right < n
        right_c = right.clamp(max=n - 1)
        valid = inb & (sent_ids.unsqueeze(1) == sent_ids[right_c])
        keep = torch.rand((n, W), device=device, generator=self.gen) \
            < (opt.window - offs + 1).float() / opt.window
        m = (valid & keep).reshape(-1)
        li = left.expand(n, W).reshape(-1)[m]
        ri = right.reshape(-1)[m]
        if li.numel() == 0:
            return empty
        # center at i, context at i+o  AND  center at i+o, context at i
        wl, wr = words[li], words[ri]
        centers = torch.cat([wl, wr])
        contexts = torch.cat([wr, wl])
        # skip-gram: one group per pair, input = context word
        g = centers.numel()
        in_off = torch.arange(g + 1, dtype=torch.int32, device=device)
        return contexts, in_off, centers

    def build_groups(self, words: torch.Tensor, sent_ids: torch.Tensor):
        """Returns (in_idx, in_off, out_idx, out_label, out_off) with
        GLOBAL word/node ids; groups follow word2vec semantics:
        skip-gram: input = context word, outputs = center (+negs/path);
        CBOW: inputs = context words of a center, outputs = center (+...).
        """
        in_idx, in_off, centers = self.build_pairs(words, sent_ids)
        if centers.numel() == 0:
            e = torch.empty(0, dtype=torch.int64, device=self.device)
            z = torch.zeros(1, dtype=torch.int32, device=self.device)
            return e, z, e, e.float(), z
        out_idx, out_label, out_off = self._outputs_for(centers)
        return in_idx, in_off, out_idx, out_label, out_off

    def _build_cbow_pairs(self, words: torch.Tensor,
                           sent_ids: torch.Tensor):
        opt = self.opt
        device = self.device
        n = words.numel()
        W = min(opt.window, n - 1)
        empty = (torch.empty(0, dtype=torch.int64, device=device),
                 torch.zeros(1, dtype=torch.int32, device=device),
                 torch.empty(0, dtype=torch.int64, device=device))
        if W < 1:
            return empty
        offs = torch.arange(1, W + 1, device=device)
        left = torch.arange(n, device=device).unsqueeze(1)
        right = left + offs
        inb = right < n
        right_c = right.clamp(max=n - 1)
        valid = inb & (sent_ids.unsqueeze(1) == sent_ids[right_c])
        keep = torch.rand((n, W), device=device, generator=self.gen) \
            < (opt.window - offs + 1).float() / opt.window
        m = (valid & keep).reshape(-1)
        li = left.expand(n, W).reshape(-1)[m]
        ri = right.reshape(-1)[m]
        if li.numel() == 0:
            return empty
        pos = torch.cat([li, ri])      # center positions
        ctx = torch.cat([words[ri], words[li]])
        order = torch.argsort(pos, stable=True)
        pos, ctx = pos[order], ctx[order]
        upos, counts = torch.unique_consecutive(pos, return_counts=True)
        in_off = torch.zeros(upos.numel() + 1, dtype=torch.int32,
                             device=device)
        in_off[1:] = counts.cumsum(0).to(torch.int32)
        return ctx, in_off, words[upos]

    def _outputs_for(self, centers: torch.Tensor):
        """Output node/label lists per group for NS and/or HS. Negatives
        draw from the per-block pool (reference PrepareData,
        wordembedding.cpp:190-209: negative_num x |unique inputs| samples
        per block) — this bounds the block's touched output rows."""
        opt = self.opt
        device = self.device
        g = centers.numel()
        idx_parts, label_parts, len_parts = [], [], []
        if opt.negative_num > 0:
            pool = self._block_pool
            if pool is not None and pool.numel() > 0:
                pick = torch.randint(0, pool.numel(),
                                     (g, opt.negative_num), device=device,
                                     generator=self.gen)
                negs = pool[pick]
            else:
                negs = self.sampler.negative_sampling(
                    (g, opt.negative_num), generator=self.gen)
            outs = torch.cat([centers.unsqueeze(1), negs], dim=1)
            labels = torch.zeros(g, 1 + opt.negative_num, device=device)
            labels[:, 0] = 1.0
            # a negative that collides with the positive target is dropped,
            # not trained with label 0 (wordembedding.cpp:279 `continue`)
            keep = torch.ones_like(outs, dtype=torch.bool)
            keep[:, 1:] = negs != centers.unsqueeze(1)
            idx_parts.append((outs, keep))
            label_parts.append(labels)
            len_parts.append(keep.sum(1))
        if opt.hs:
            starts = self._hs_off[centers]
            lens = self._hs_off[centers + 1] - starts
            flat_pos = (starts.repeat_interleave(lens)
                        + _segment_arange(lens))
            hs_idx = self._hs_point[flat_pos]
            hs_lab = self._hs_label[flat_pos]
            if idx_parts:
                # NS + HS together: interleave per group
                (ns_idx, ns_keep), ns_lab = idx_parts[0], label_parts[0]
                ns_lens = len_parts[0]
                out_lens = lens + ns_lens
                off = _lens_to_off(out_lens)
                total = int(out_lens.sum())
                out_idx = torch.empty(total, dtype=torch.int64, device=device)
                out_lab = torch.empty(total, device=device)
                ns_pos = (off[:-1].repeat_interleave(ns_lens)
                          + _segment_arange(ns_lens))
                out_idx[ns_pos] = ns_idx[ns_keep]
                out_lab[ns_pos] = ns_lab[ns_keep]
                hs_pos = ((off[:-1] + ns_lens).repeat_interleave(lens)
                          + _segment_arange(lens))
                out_idx[hs_pos] = hs_idx
                out_lab[hs_pos] = hs_lab
                return out_idx, out_lab, off.to(torch.int32)
            off = _lens_to_off(lens)
            return hs_idx, hs_lab, off.to(torch.int32)
        (ns_idx, ns_keep), ns_lab = idx_parts[0], label_parts[0]
        off = _lens_to_off(len_parts[0])
        return ns_idx[ns_keep], ns_lab[ns_keep], off.to(torch.int32)

    # ------------------------------------------------------------------
    # Block training
    # ------------------------------------------------------------------
    def train_block(self, words: torch.Tensor,
                    sent_ids: torch.Tensor) -> int:
        """Train one data block; returns words processed. The caller is
        responsible for calling this collectively on every rank (the
        block pipeline of distributed_wordembedding.cpp:147-252)."""
        opt = self.opt
        words = words.to(self.device)
        sent_ids = sent_ids.to(self.device)
        if opt.sample > 0:
            m = self.sampler.keep_mask(words, opt.sample, self.gen)
            words, sent_ids = words[m], sent_ids[m]
        uwords = torch.unique(words)
        self._block_pool = None
        if opt.negative_num > 0:
            # deduped, like the reference's negativesample_pools std::set
            # (wordembedding.cpp:208)
            self._block_pool = torch.unique(self.sampler.negative_sampling(
                (opt.negative_num * uwords.numel(),), generator=self.gen))

        # NS fast path (GPU, negative sampling without HS): negatives are
        # generated INSIDE the kernel from the block pool with the
        # reference's LCG scheme (wordembedding.cpp:276-279) — no
        # host-side randint/gather/label/ragged build at all.
        ns_fast = (opt.negative_num > 0 and not opt.hs
                   and self.device.type == "cuda")
        empty_block = False
        if ns_fast:
            in_idx, in_off, centers = self.build_pairs(words, sent_ids)
            empty_block = centers.numel() == 0
            out_idx = out_label = out_off = None
        else:
            in_idx, in_off, out_idx, out_label, out_off = \
                self.build_groups(words, sent_ids)
            empty_block = in_idx.numel() == 0 or out_idx.numel() == 0
        if empty_block and mv.size() == 1:
            return int(words.numel())
        # multi-rank: an empty block still participates in the collective
        # pull/push below with zero-sized exchanges — ranks can hold
        # different block counts (uneven corpus stripes) as long as the
        # driver loop keeps calling collectively (main.py).

        if mv.size() == 1:
            # Single-rank fast path: every row is local, and
            # pull → train → push(trained − pulled) is identical to
            # training the HBM-resident shard in place (P=1). Skips two
            # unique-sorts, the gathers/scatters and the delta pass.
            igq = ogq = None
            if opt.use_adagrad:
                igq = self.input_gsq_table.shard
                ogq = self.output_gsq_table.shard
            if ns_fast:
                self._train_kernel_ns(self.input_table.shard,
                                      self.output_table.shard, igq, ogq,
                                      in_idx,
                                      in_off if opt.cbow else None,
                                      centers, self._block_pool)
            else:
                self._train_kernel(self.input_table.shard,
                                   self.output_table.shard, igq, ogq,
                                   in_idx, in_off, out_idx,
                                   out_label.float(), out_off)
            nwords = int(words.numel())
            self._update_lr(nwords)
            return nwords

        # Touched-row candidate sets are known up front (block vocabulary
        # + negative pool + HS path nodes), so the 10M+-element
        # unique-sorts collapse to searchsorted over small sorted sets.
        uin = uwords  # inputs are always block words
        in_local = torch.searchsorted(uin, in_idx)
        cand = [uwords] if (opt.negative_num > 0 or not opt.hs) else []
        if self._block_pool is not None:
            cand.append(self._block_pool)
        if opt.hs:
            starts = self._hs_off[uwords]
            lens = self._hs_off[uwords + 1] - starts
            cand.append(self._hs_point[
                starts.repeat_interleave(lens) + _segment_arange(lens)])
        uout = torch.unique(torch.cat(cand))
        if ns_fast:
            out_local = torch.searchsorted(uout, centers)
            pool_local = torch.searchsorted(uout, self._block_pool)
        else:
            out_local = torch.searchsorted(uout, out_idx)

        # pull touched rows (RequestParameter)
        in_buf = self.input_table.get_rows(uin).contiguous()
        out_buf = self.output_table.get_rows(uout).contiguous()
        in_old = in_buf.clone()
        out_old = out_buf.clone()
        gbufs = (None, None, None, None)
        if opt.use_adagrad:
            igq = self.input_gsq_table.get_rows(uin).contiguous()
            ogq = self.output_gsq_table.get_rows(uout).contiguous()
            gbufs = (igq, igq.clone(), ogq, ogq.clone())

        if ns_fast:
            if out_local.numel() or in_local.numel():
                self._train_kernel_ns(in_buf, out_buf, gbufs[0], gbufs[2],
                                      in_local,
                                      in_off if opt.cbow else None,
                                      out_local, pool_local)
        elif out_local.numel():
            self._train_kernel(in_buf, out_buf, gbufs[0], gbufs[2],
                               in_local, in_off, out_local,
                               out_label.float(), out_off)

        # push deltas (AddDeltaParameter, /num_workers)
        p = float(mv.workers_num())
        self.input_table.add_rows(uin, (in_buf - in_old) / p)
        self.output_table.add_rows(uout, (out_buf - out_old) / p)
        if opt.use_adagrad:
            self.input_gsq_table.add_rows(uin, (gbufs[0] - gbufs[1]) / p)
            self.output_gsq_table.add_rows(uout, (gbufs[2] - gbufs[3]) / p)

        nwords = int(words.numel())
        self._update_lr(nwords)
        return nwords

    def _next_seed(self) -> int:
        """Per-launch seed for the in-kernel negative LCG: a host-side
        splitmix step — no device sync, reproducible from opt.seed."""
        s = (self._seed_state + 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF
        self._seed_state = s
        z = s
        z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
        z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
        return (z ^ (z >> 31)) & 0x7FFFFFFFFFFFFFFF

    def _train_kernel_ns(self, in_buf, out_buf, in_gsq, out_gsq,
                         in_local, in_off, centers, pool):
        """NS fast path: negatives generated in-kernel from `pool`.
        ``in_off=None`` = skip-gram (exactly one input per group; the
        kernel skips the offset loads — probe V7)."""
        from ... import ops
        hip = ops.module(required=True)
        dummy = in_buf
        igq = in_gsq if in_gsq is not None else dummy
        ogq = out_gsq if out_gsq is not None else dummy
        if in_off is not None:
            in_off = in_off.to(torch.int32)
        g_total = centers.numel()
        step = self.opt.max_groups_per_launch
        if g_total <= step:
            hip.w2v_train_ns(in_buf, out_buf, igq, ogq, in_local, in_off,
                             centers, pool, self.opt.negative_num,
                             self._next_seed(), self.learning_rate,
                             self.opt.use_adagrad, self.opt.atomic_updates)
            return
        gs = list(range(0, g_total, step)) + [g_total]
        ibs = gs if in_off is None else in_off[gs].tolist()
        for k in range(len(gs) - 1):
            g0, g1 = gs[k], gs[k + 1]
            i0, i1 = ibs[k], ibs[k + 1]
            hip.w2v_train_ns(in_buf, out_buf, igq, ogq,
                             in_local[i0:i1].contiguous(),
                             None if in_off is None else
                             (in_off[g0:g1 + 1] - i0).contiguous(),
                             centers[g0:g1].contiguous(), pool,
                             self.opt.negative_num, self._next_seed(),
                             self.learning_rate, self.opt.use_adagrad,
                             self.opt.atomic_updates)

    def _train_kernel(self, in_buf, out_buf, in_gsq, out_gsq,
                      in_local, in_off, out_local, out_label, out_off):
        if in_buf.is_cuda:
            from ... import ops
            hip = ops.module(required=True)
            dummy = in_buf  # unused when not adagrad
            igq = in_gsq if in_gsq is not None else dummy
            ogq = out_gsq if out_gsq is not None else dummy
            in_off = in_off.to(torch.int32)
            out_off = out_off.to(torch.int32)
            g_total = in_off.numel() - 1
            step = self.opt.max_groups_per_launch
            if g_total <= step:
                # single launch: no slicing, no host sync
                hip.w2v_train(in_buf, out_buf, igq, ogq,
                              in_local, in_off, out_local, out_label,
                              out_off, self.learning_rate,
                              self.opt.use_adagrad,
                              self.opt.atomic_updates)
                return
            # one host transfer for every chunk boundary (int(in_off[g])
            # per chunk would sync the stream 4x per chunk)
            gs = list(range(0, g_total, step)) + [g_total]
            ibs = in_off[gs].tolist()
            obs = out_off[gs].tolist()
            for k in range(len(gs) - 1):
                g0, g1 = gs[k], gs[k + 1]
                i0, i1 = ibs[k], ibs[k + 1]
                o0, o1 = obs[k], obs[k + 1]
                hip.w2v_train(in_buf, out_buf, igq, ogq,
                              in_local[i0:i1].contiguous(),
                              (in_off[g0:g1 + 1] - i0).contiguous(),
                              out_local[o0:o1].contiguous(),
                              out_label[o0:o1].contiguous(),
                              (out_off[g0:g1 + 1] - o0).contiguous(),
                              self.learning_rate, self.opt.use_adagrad,
                              self.opt.atomic_updates)
        else:
            _w2v_train_torch(in_buf, out_buf, in_gsq, out_gsq, in_local,
                             in_off, out_local, out_label, out_off,
                             self.learning_rate, self.opt.use_adagrad,
                             self.opt.init_learning_rate)

    def _update_lr(self, nwords: int) -> None:
        """Accumulate local progress; the decay itself is applied by
        sync_word_count() from the global count (wordembedding.cpp:36-47)."""
        self.word_count_local += nwords

    def sync_word_count(self) -> None:
        """Collective word-count sync via the KV table (AddDeltaWordCount,
        distributed_wordembedding.cpp:82-90)."""
        self.word_count_table.add([0], [float(self.word_count_local)])
        self.word_count_local = 0
        self.word_count_actual = int(self.word_count_table.get([0])[0])
        opt = self.opt
        denom = float(opt.total_words * opt.epoch) + 1.0
        lr = opt.init_learning_rate * (1 - self.word_count_actual / denom)
        self.learning_rate = max(lr, opt.init_learning_rate * 1e-4)

    # ------------------------------------------------------------------
    def save_embedding(self, path: str, words: List[str],
                       binary: bool = False) -> None:
        """word2vec-format save (SaveEmbedding,
        distributed_wordembedding.cpp:263-306); rank 0 writes, pulling in
        chunks of kSaveBatch=100000 rows."""
        dim = self.opt.embedding_size
        batch = 100_000
        if mv.rank() == 0:
            f = open(path, "wb" if binary else "w")
            header = f"{self.vocab_size} {dim}\n"
            f.write(header.encode() if binary else header)
        mv.barrier()
        for start in range(0, self.vocab_size, batch):
            ids = list(range(start, min(start + batch, self.vocab_size)))
            rows = self.input_table.get_rows(ids)
            if mv.rank() == 0:
                rows_cpu = rows.cpu().numpy()
                for i, wid in enumerate(ids):
                    if binary:
                        f.write((words[wid] + " ").encode())
                        f.write(rows_cpu[i].tobytes())
                        f.write(b"\n")
                    else:
                        vec = " ".join(f"{x:.6f}" for x in rows_cpu[i])
                        f.write(f"{words[wid]} {vec}\n")
        if mv.rank() == 0:
            f.close()
        mv.barrier()


# ---------------------------------------------------------------------------
# CPU reference implementation of the fused kernel (test oracle + fallback)
# ---------------------------------------------------------------------------

def _w2v_train_torch(in_buf, out_buf, in_gsq, out_gsq, in_idx, in_off,
                     out_idx, out_label, out_off, lr, use_adagrad, lr0):
    """Sequential group-by-group reference of k_w2v (same math as
    wordembedding.cpp:57-166)."""
    g_count = in_off.numel() - 1
    in_off_l = in_off.tolist()
    out_off_l = out_off.tolist()
    for g in range(g_count):
        ii = in_idx[in_off_l[g]:in_off_l[g + 1]]
        oo = out_idx[out_off_l[g]:out_off_l[g + 1]]
        ll = out_label[out_off_l[g]:out_off_l[g + 1]]
        h = in_buf[ii].mean(dim=0)
        err = torch.zeros_like(h)
        for j in range(oo.numel()):
            r = int(oo[j])
            w = out_buf[r]  # row view -> in-place updates write through
            f = torch.sigmoid(h @ w)
            e = float(ll[j]) - float(f)
            err += e * w
            if use_adagrad:
                gg = e * h
                gq = out_gsq[r]
                gq += gg * gg
                mask = gq > 1e-10
                w[mask] += (gg * lr0 / torch.sqrt(gq))[mask]
            else:
                w += e * lr * h
        for i in range(ii.numel()):
            r = int(ii[i])
            if use_adagrad:
                gq = in_gsq[r]
                gq += err * err
                mask = gq > 1e-10
                in_buf[r][mask] += (err * lr0 / torch.sqrt(gq))[mask]
            else:
                in_buf[r] += lr * err


def _segment_arange(lens: torch.Tensor) -> torch.Tensor:
    """[0..l0-1, 0..l1-1, ...] for a tensor of segment lengths."""
    total = int(lens.sum())
    if total == 0:
        return torch.empty(0, dtype=torch.int64, device=lens.device)
    off = torch.zeros_like(lens)
    off[1:] = lens.cumsum(0)[:-1]
    return (torch.arange(total, device=lens.device)
            - off.repeat_interleave(lens))


def _lens_to_off(lens: torch.Tensor) -> torch.Tensor:
    off = torch.zeros(lens.numel() + 1, dtype=torch.int64,
                      device=lens.device)
    off[1:] = lens.cumsum(0)
    return off
