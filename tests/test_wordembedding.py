"""WordEmbedding app tests (CPU): dictionary, huffman, sampler, group
construction invariants, and an end-to-end tiny training run whose loss
must decrease (the functional oracle for the fused path)."""

import numpy as np
import pytest
import torch

import multiverso_amd as mv
from multiverso_amd.apps.wordembedding import (Dictionary, HuffmanEncoder,
                                               Sampler, WordEmbedding,
                                               WordEmbeddingOption)
from multiverso_amd.apps.wordembedding.data import (synthetic_block,
                                                    zipf_counts)


@pytest.fixture()
def env():
    mv.init()
    yield
    mv.shutdown()


def test_dictionary_roundtrip(tmp_path):
    d = Dictionary.build("a b c a b a".split(), min_count=1)
    assert d.get_id("a") == 0  # most frequent first
    assert d.get_count(d.get_id("a")) == 3
    p = str(tmp_path / "vocab.txt")
    d.save(p)
    d2 = Dictionary.load(p)
    assert len(d2) == 3 and d2.get_count(d2.get_id("b")) == 2


def test_dictionary_min_count():
    d = Dictionary.build("a a a b".split(), min_count=2)
    assert d.get_id("b") == -1 and len(d) == 1


def test_huffman_codes_prefix_free():
    enc = HuffmanEncoder()
    counts = [50, 30, 10, 5, 3, 2]
    enc.build_from_term_frequency(counts)
    codes = ["".join(map(str, enc.get_label_info(i).code))
             for i in range(len(counts))]
    # prefix-free and frequent words get shorter codes
    for i, a in enumerate(codes):
        for j, b in enumerate(codes):
            if i != j:
                assert not b.startswith(a)
    assert len(codes[0]) <= len(codes[-1])
    # point ids are inner nodes in [0, V-1)
    for i in range(len(counts)):
        info = enc.get_label_info(i)
        assert len(info.point) == len(info.code)
        assert all(0 <= p < len(counts) - 1 for p in info.point)


def test_sampler_distribution():
    counts = [1000, 100, 10, 1]
    s = Sampler(counts, table_size=100_000)
    draws = s.negative_sampling((50_000,))
    freq = torch.bincount(draws, minlength=4).float()
    assert freq[0] > freq[1] > freq[2] > freq[3] > 0


def test_group_building_skipgram(env):
    opt = WordEmbeddingOption(embedding_size=16, window=2, negative_num=3,
                              total_words=1000)
    model = WordEmbedding(opt, [100] * 50)
    words = torch.arange(20, dtype=torch.int64)
    sids = torch.zeros(20, dtype=torch.int64)
    in_idx, in_off, out_idx, out_label, out_off = \
        model.build_groups(words, sids)
    g = in_off.numel() - 1
    assert g > 0
    assert out_off.numel() - 1 == g
    # 1 pos + up to 3 negs per group (negatives colliding with the target
    # are dropped, wordembedding.cpp:279)
    lens = (out_off[1:] - out_off[:-1]).long()
    assert torch.all(lens >= 1) and torch.all(lens <= 4)
    assert int(lens.sum()) == int(out_off[-1])
    first = out_off[:-1].long()
    assert torch.all(out_label[first] == 1)  # positive leads each group
    neg_mask = torch.ones(out_idx.numel(), dtype=torch.bool)
    neg_mask[first] = False
    assert torch.all(out_label[neg_mask] == 0)
    # negatives never equal their group's positive target
    pos = out_idx[first]
    gid = torch.repeat_interleave(torch.arange(g), lens)
    assert torch.all(out_idx[neg_mask] != pos[gid[neg_mask]])
    # positives are within window distance of their input
    assert torch.all((pos - in_idx).abs() <= opt.window)


def test_group_building_cbow_hs(env):
    opt = WordEmbeddingOption(embedding_size=16, window=2, negative_num=0,
                              hs=True, cbow=True, total_words=1000)
    model = WordEmbedding(opt, [100, 90, 80, 70, 60, 50, 40, 30])
    words = torch.tensor([0, 1, 2, 3, 4, 5, 6, 7])
    sids = torch.zeros(8, dtype=torch.int64)
    in_idx, in_off, out_idx, out_label, out_off = \
        model.build_groups(words, sids)
    g = in_off.numel() - 1
    assert g > 0
    assert torch.all(out_label.ge(0) & out_label.le(1))
    # hs points are inner nodes
    assert int(out_idx.max()) < 7


def _avg_pair_logit(model, pairs):
    inp = model.input_table.get()
    out = model.output_table.get()
    return float(torch.sigmoid((inp[pairs[:, 0]] * out[pairs[:, 1]])
                               .sum(1)).mean())


def test_train_block_learns(env):
    """After training on a corpus where word 2k is always followed by
    2k+1, observed pairs must score far higher than mismatched pairs."""
    torch.manual_seed(0)
    vocab = 20
    opt = WordEmbeddingOption(embedding_size=16, window=1, negative_num=5,
                              init_learning_rate=0.1,
                              total_words=10_000_000, seed=3)
    model = WordEmbedding(opt, [100] * vocab)
    words = torch.stack([torch.arange(0, 20, 2).repeat(50),
                         torch.arange(1, 20, 2).repeat(50)], dim=1).view(-1)
    perm = torch.randperm(words.numel() // 2)
    words = words.view(-1, 2)[perm].view(-1)
    sids = torch.arange(words.numel()) // 10
    for i in range(10):
        nw = model.train_block(words, sids)
        assert nw == words.numel()
        model.sync_word_count()
    evens = torch.arange(0, 20, 2)
    pos = _avg_pair_logit(model, torch.stack([evens, evens + 1], dim=1))
    wrong = _avg_pair_logit(model, torch.stack(
        [evens, evens.roll(1) + 1], dim=1))
    assert pos > 0.6 and wrong < 0.2, (pos, wrong)
    assert model.word_count_actual == 10 * words.numel()
    assert model.learning_rate < opt.init_learning_rate


def test_save_embedding(env, tmp_path):
    opt = WordEmbeddingOption(embedding_size=8, total_words=100)
    model = WordEmbedding(opt, [10] * 20)
    vocab_words = [f"w{i}" for i in range(20)]
    p = str(tmp_path / "emb.txt")
    model.save_embedding(p, vocab_words)
    lines = open(p).read().splitlines()
    assert lines[0] == "20 8"
    assert len(lines) == 21
    assert lines[1].split()[0] == "w0"
    assert len(lines[1].split()) == 9


def _w2v_dist(rank, world):
    import torch
    import multiverso_amd as mv
    from multiverso_amd.apps.wordembedding.model import (WordEmbedding,
                                                         WordEmbeddingOption)
    mv.init(sync=True)
    torch.manual_seed(rank)
    opt = WordEmbeddingOption(embedding_size=16, window=1, negative_num=3,
                              init_learning_rate=0.1,
                              total_words=10_000_000, seed=5)
    model = WordEmbedding(opt, [100] * 12)
    words = torch.stack([torch.arange(0, 12, 2).repeat(30),
                         torch.arange(1, 12, 2).repeat(30)], dim=1).view(-1)
    sids = torch.arange(words.numel()) // 10
    for _ in range(6):
        model.train_block(words, sids)       # multi-rank pull/push path
        model.sync_word_count()
    assert model.word_count_actual == 6 * words.numel() * world
    inp = model.input_table.get()
    out = model.output_table.get()
    evens = torch.arange(0, 12, 2)
    pos = torch.sigmoid((inp[evens] * out[evens + 1]).sum(1)).mean()
    wrong = torch.sigmoid((inp[evens] * out[evens.roll(1) + 1]).sum(1)).mean()
    assert float(pos) - float(wrong) > 0.2, (rank, float(pos), float(wrong))
    mv.shutdown()


def test_w2v_dist_two_ranks():
    from conftest import run_dist
    run_dist(_w2v_dist, 2)


def _w2v_uneven_blocks(rank, world):
    """Uneven per-rank block counts: rank 0 trains 2 blocks, rank 1
    trains 1 — the collective loop (empty-block participation) must not
    desynchronize. Regression test for the multi-rank CLI hang."""
    import torch
    import multiverso_amd as mv
    from multiverso_amd.apps.wordembedding.model import (WordEmbedding,
                                                         WordEmbeddingOption)
    mv.init()
    opt = WordEmbeddingOption(embedding_size=16, window=1, negative_num=2,
                              total_words=1000, seed=4)
    model = WordEmbedding(opt, [50] * 12)
    g = torch.Generator().manual_seed(rank)
    blocks = [(torch.randint(0, 12, (40,), generator=g),
               torch.arange(40) // 8) for _ in range(2 if rank == 0 else 1)]
    empty = (torch.empty(0, dtype=torch.int64),
             torch.empty(0, dtype=torch.int64))
    it = iter(blocks)
    while True:
        blk = next(it, None)
        have = torch.tensor([0.0 if blk is None else 1.0])
        mv.aggregate(have)
        if float(have[0]) == 0.0:
            break
        model.train_block(*(blk if blk is not None else empty))
        model.sync_word_count()
    # default mode is now TRUE ASYNC: the global count is only exact
    # after a drain barrier (mid-training reads may lag — by design)
    mv.barrier()
    model.sync_word_count()
    assert model.word_count_actual == 120  # 2*40 + 1*40
    mv.shutdown()


def test_w2v_uneven_blocks_dist():
    from conftest import run_dist
    run_dist(_w2v_uneven_blocks, 2)


def test_huffman_exact_reference_layout():
    """Exact point/code arrays for a hand-traced run of the reference's
    two-queue construction (huffman_encoder.cpp:87-188) on counts
    [4,2,1,1]: code = root->leaf branch labels, point = inner-node path
    (root = V-2 first, leaf's parent last)."""
    enc = HuffmanEncoder()
    enc.build_from_term_frequency([4, 2, 1, 1])
    expect = [([2], [1]),
              ([2, 1], [0, 1]),
              ([2, 1, 0], [0, 0, 1]),
              ([2, 1, 0], [0, 0, 0])]
    for i, (point, code) in enumerate(expect):
        assert enc.labels[i].point == point, (i, enc.labels[i].point)
        assert enc.labels[i].code == code, (i, enc.labels[i].code)


def test_huffman_file_roundtrip(tmp_path):
    """Save2File/RecoverFromFile format (huffman_encoder.cpp:9-85)."""
    enc = HuffmanEncoder()
    enc.build_from_term_frequency([9, 5, 3, 2, 1])
    p = str(tmp_path / "huff.txt")
    enc.save_to_file(p, ["a", "b", "c", "d", "e"])
    enc2 = HuffmanEncoder()
    words = enc2.load_from_file(p)
    assert words == ["a", "b", "c", "d", "e"]
    for x, y in zip(enc.labels, enc2.labels):
        assert x.code == y.code and x.point == y.point
    first = open(p).read().splitlines()
    assert first[0] == "5" and first[1].startswith("a ")


def _w2v_hogwild_ws8(rank, world):
    """8-rank convergence oracle for the pull-train-push staleness x
    hogwild interaction: 8 workers train the same
    even->odd structure concurrently in TRUE ASYNC mode (keyed get/add
    served on arrival); after the drain barrier the embeddings must
    still separate positives from mismatched pairs."""
    import torch
    import multiverso_amd as mv
    from multiverso_amd.apps.wordembedding.model import (WordEmbedding,
                                                         WordEmbeddingOption)
    mv.init()   # async mode — the real ASGD w2v
    torch.manual_seed(rank)
    opt = WordEmbeddingOption(embedding_size=16, window=1, negative_num=3,
                              init_learning_rate=0.08,
                              total_words=50_000_000, seed=3 + rank)
    model = WordEmbedding(opt, [100] * 12)
    words = torch.stack([torch.arange(0, 12, 2).repeat(20),
                         torch.arange(1, 12, 2).repeat(20)], dim=1).view(-1)
    sids = torch.arange(words.numel()) // 10
    for _ in range(4):
        model.train_block(words, sids)
        model.sync_word_count()
    mv.barrier()
    inp = model.input_table.get()
    out = model.output_table.get()
    evens = torch.arange(0, 12, 2)
    pos = torch.sigmoid((inp[evens] * out[evens + 1]).sum(1)).mean()
    wrong = torch.sigmoid((inp[evens] * out[evens.roll(1) + 1]).sum(1)).mean()
    assert float(pos) - float(wrong) > 0.15, (rank, float(pos), float(wrong))
    mv.shutdown()


def test_w2v_hogwild_convergence_ws8():
    from conftest import run_dist
    run_dist(_w2v_hogwild_ws8, 8, timeout=240)
