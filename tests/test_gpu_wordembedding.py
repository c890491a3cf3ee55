"""GPU tests for the fused word2vec kernel: numerics vs the sequential
torch fp32 reference (disjoint rows so group order cannot matter), and an
end-to-end learning check on GPU."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _make_disjoint_case(g=8, nin=1, nout=4, dim=200, seed=0):
    """Groups touch disjoint input and output rows -> kernel result must
    match the sequential reference to fp32 tolerance."""
    torch.manual_seed(seed)
    in_rows, out_rows = g * nin, g * nout
    in_buf = torch.randn(in_rows, dim) * 0.1
    out_buf = torch.randn(out_rows, dim) * 0.1
    in_idx = torch.arange(in_rows, dtype=torch.int64)
    in_off = torch.arange(0, in_rows + 1, nin, dtype=torch.int32)
    out_idx = torch.arange(out_rows, dtype=torch.int64)
    out_off = torch.arange(0, out_rows + 1, nout, dtype=torch.int32)
    out_label = (torch.arange(out_rows) % nout == 0).float()
    return in_buf, out_buf, in_idx, in_off, out_idx, out_label, out_off


@pytest.mark.parametrize("dim", [64, 200, 128 + 17])
@pytest.mark.parametrize("nin", [1, 3])
def test_w2v_kernel_vs_reference(dim, nin):
    from multiverso_amd import ops
    from multiverso_amd.apps.wordembedding.model import _w2v_train_torch
    hip = ops.module(required=True)
    case = _make_disjoint_case(g=16, nin=nin, nout=5, dim=dim)
    in_buf, out_buf, in_idx, in_off, out_idx, out_label, out_off = case
    lr = 0.05

    ref_in, ref_out = in_buf.clone(), out_buf.clone()
    _w2v_train_torch(ref_in, ref_out, None, None, in_idx, in_off,
                     out_idx, out_label, out_off, lr, False, lr)

    d = lambda t: t.cuda()
    gin, gout = d(in_buf), d(out_buf)
    hip.w2v_train(gin, gout, gin, gout, d(in_idx), d(in_off), d(out_idx),
                  d(out_label), d(out_off), lr, False, False)
    torch.cuda.synchronize()
    assert torch.allclose(gin.cpu(), ref_in, rtol=1e-4, atol=1e-5), \
        (gin.cpu() - ref_in).abs().max()
    assert torch.allclose(gout.cpu(), ref_out, rtol=1e-4, atol=1e-5)


def test_w2v_kernel_adagrad_vs_reference():
    from multiverso_amd import ops
    from multiverso_amd.apps.wordembedding.model import _w2v_train_torch
    hip = ops.module(required=True)
    case = _make_disjoint_case(g=8, nin=2, nout=4, dim=96, seed=1)
    in_buf, out_buf, in_idx, in_off, out_idx, out_label, out_off = case
    in_gsq = torch.zeros_like(in_buf)
    out_gsq = torch.zeros_like(out_buf)
    lr = 0.05

    ref = [in_buf.clone(), out_buf.clone(), in_gsq.clone(), out_gsq.clone()]
    _w2v_train_torch(ref[0], ref[1], ref[2], ref[3], in_idx, in_off,
                     out_idx, out_label, out_off, lr, True, lr)

    d = lambda t: t.cuda()
    g = [d(in_buf), d(out_buf), d(in_gsq), d(out_gsq)]
    hip.w2v_train(g[0], g[1], g[2], g[3], d(in_idx), d(in_off), d(out_idx),
                  d(out_label), d(out_off), lr, True, False)
    torch.cuda.synchronize()
    for got, want in zip(g, ref):
        assert torch.allclose(got.cpu(), want, rtol=1e-3, atol=1e-5), \
            (got.cpu() - want).abs().max()


def test_w2v_ns_kernel_deterministic_pool_vs_reference():
    """NS fast path with a 1-element pool and a single group: every
    in-kernel negative is the known pool row, so the output sequence is
    exactly (positive, neg x pool-row) and — with one wave and plain
    stores (own-thread coherent) — the kernel must match the sequential
    torch reference to fp32 tolerance."""
    from multiverso_amd import ops
    from multiverso_amd.apps.wordembedding.model import _w2v_train_torch
    hip = ops.module(required=True)
    dim, neg = 128, 3
    torch.manual_seed(4)
    in_buf = torch.randn(1, dim) * 0.1
    out_buf = torch.randn(2, dim) * 0.1
    in_idx = torch.zeros(1, dtype=torch.int64)
    in_off = torch.tensor([0, 1], dtype=torch.int32)
    centers = torch.zeros(1, dtype=torch.int64)     # positive = out row 0
    pool = torch.tensor([1], dtype=torch.int64)     # negatives = out row 1
    lr = 0.05

    ref_in, ref_out = in_buf.clone(), out_buf.clone()
    out_idx = torch.tensor([0] + [1] * neg, dtype=torch.int64)
    lab = torch.tensor([1.0] + [0.0] * neg)
    out_off = torch.tensor([0, 1 + neg], dtype=torch.int32)
    _w2v_train_torch(ref_in, ref_out, None, None, in_idx, in_off,
                     out_idx, lab, out_off, lr, False, lr)

    d = lambda t: t.cuda()
    gin, gout = d(in_buf), d(out_buf)
    hip.w2v_train_ns(gin, gout, gin, gout, d(in_idx), d(in_off), d(centers),
                     d(pool), neg, 12345, lr, False, False)
    torch.cuda.synchronize()
    assert torch.allclose(gin.cpu(), ref_in, rtol=1e-4, atol=1e-5), \
        (gin.cpu() - ref_in).abs().max()
    assert torch.allclose(gout.cpu(), ref_out, rtol=1e-4, atol=1e-5), \
        (gout.cpu() - ref_out).abs().max()


def test_w2v_ns_kernel_only_touches_pool_and_centers():
    """In-kernel negatives must come from the pool: rows outside
    centers ∪ pool stay bit-identical."""
    from multiverso_amd import ops
    hip = ops.module(required=True)
    g, dim, V = 8, 64, 64
    torch.manual_seed(5)
    in_buf = torch.randn(V, dim).cuda() * 0.1
    out_buf = torch.randn(V, dim).cuda() * 0.1
    before = out_buf.clone()
    in_idx = torch.arange(g, dtype=torch.int64).cuda()
    in_off = torch.arange(g + 1, dtype=torch.int32).cuda()
    centers = torch.arange(10, 10 + g, dtype=torch.int64).cuda()
    pool = torch.tensor([40, 41, 42], dtype=torch.int64).cuda()
    hip.w2v_train_ns(out_buf, out_buf, out_buf, out_buf, in_idx, in_off,
                     centers, pool, 5, 999, 0.1, False, False)
    torch.cuda.synchronize()
    touched = set(range(g)) | set(range(10, 10 + g)) | {40, 41, 42}
    untouched = [r for r in range(V) if r not in touched]
    assert torch.equal(out_buf[untouched], before[untouched])
    assert not torch.equal(out_buf[list(touched)], before[list(touched)])


def test_w2v_gpu_end_to_end_learns():
    import multiverso_amd as mv
    from multiverso_amd.apps.wordembedding.model import (WordEmbedding,
                                                         WordEmbeddingOption)
    mv.init()
    torch.manual_seed(0)
    # duplicate-heavy corpus: chunk launches small so later duplicates see
    # earlier updates (sequential-equivalent semantics)
    opt = WordEmbeddingOption(embedding_size=64, window=1, negative_num=5,
                              init_learning_rate=0.1,
                              total_words=10_000_000, seed=3,
                              max_groups_per_launch=64)
    model = WordEmbedding(opt, [100] * 20)
    words = torch.stack([torch.arange(0, 20, 2).repeat(100),
                         torch.arange(1, 20, 2).repeat(100)],
                        dim=1).view(-1).cuda()
    sids = (torch.arange(words.numel()) // 10).cuda()
    for _ in range(10):
        model.train_block(words, sids)
    inp = model.input_table.get()
    out = model.output_table.get()
    evens = torch.arange(0, 20, 2, device="cuda")
    pos = torch.sigmoid((inp[evens] * out[evens + 1]).sum(1)).mean()
    wrong = torch.sigmoid((inp[evens] * out[evens.roll(1) + 1]).sum(1)).mean()
    torch.cuda.synchronize()
    # hogwild updates lose some negative-pair pushes on this tiny
    # duplicate-heavy corpus; require strong discrimination margin
    assert float(pos) > 0.6 and float(pos) - float(wrong) > 0.3, \
        (float(pos), float(wrong))
    mv.shutdown()


def test_w2v_ns_kernel_adagrad_vs_reference():
    """NS fast path with adagrad state, single group + 1-element pool:
    exact sequential semantics vs the torch reference."""
    from multiverso_amd import ops
    from multiverso_amd.apps.wordembedding.model import _w2v_train_torch
    hip = ops.module(required=True)
    dim, neg = 96, 3
    torch.manual_seed(6)
    in_buf = torch.randn(1, dim) * 0.1
    out_buf = torch.randn(2, dim) * 0.1
    in_gsq = torch.zeros(1, dim)
    out_gsq = torch.zeros(2, dim)
    in_idx = torch.zeros(1, dtype=torch.int64)
    in_off = torch.tensor([0, 1], dtype=torch.int32)
    centers = torch.zeros(1, dtype=torch.int64)
    pool = torch.tensor([1], dtype=torch.int64)
    lr = 0.05

    ref = [in_buf.clone(), out_buf.clone(), in_gsq.clone(), out_gsq.clone()]
    out_idx = torch.tensor([0] + [1] * neg, dtype=torch.int64)
    lab = torch.tensor([1.0] + [0.0] * neg)
    out_off = torch.tensor([0, 1 + neg], dtype=torch.int32)
    _w2v_train_torch(ref[0], ref[1], ref[2], ref[3], in_idx, in_off,
                     out_idx, lab, out_off, lr, True, lr)

    d = lambda t: t.cuda()
    g = [d(in_buf), d(out_buf), d(in_gsq), d(out_gsq)]
    hip.w2v_train_ns(g[0], g[1], g[2], g[3], d(in_idx), None, d(centers),
                     d(pool), neg, 777, lr, True, False)
    torch.cuda.synchronize()
    for got, want in zip(g, ref):
        assert torch.allclose(got.cpu(), want, rtol=1e-3, atol=1e-5), \
            (got.cpu() - want).abs().max()


def test_w2v_ns_kernel_cbow_ragged_inputs():
    """NS fast path with ragged in_off (CBOW: several inputs averaged per
    group), single group, deterministic pool."""
    from multiverso_amd import ops
    from multiverso_amd.apps.wordembedding.model import _w2v_train_torch
    hip = ops.module(required=True)
    dim, neg, nin = 64, 2, 3
    torch.manual_seed(7)
    in_buf = torch.randn(nin, dim) * 0.1
    out_buf = torch.randn(2, dim) * 0.1
    in_idx = torch.arange(nin, dtype=torch.int64)
    in_off = torch.tensor([0, nin], dtype=torch.int32)
    centers = torch.zeros(1, dtype=torch.int64)
    pool = torch.tensor([1], dtype=torch.int64)
    lr = 0.05

    ref_in, ref_out = in_buf.clone(), out_buf.clone()
    out_idx = torch.tensor([0] + [1] * neg, dtype=torch.int64)
    lab = torch.tensor([1.0] + [0.0] * neg)
    out_off = torch.tensor([0, 1 + neg], dtype=torch.int32)
    _w2v_train_torch(ref_in, ref_out, None, None, in_idx, in_off,
                     out_idx, lab, out_off, lr, False, lr)

    d = lambda t: t.cuda()
    gin, gout = d(in_buf), d(out_buf)
    hip.w2v_train_ns(gin, gout, gin, gout, d(in_idx), d(in_off), d(centers),
                     d(pool), neg, 555, lr, False, False)
    torch.cuda.synchronize()
    assert torch.allclose(gin.cpu(), ref_in, rtol=1e-4, atol=1e-5)
    assert torch.allclose(gout.cpu(), ref_out, rtol=1e-4, atol=1e-5)


def test_w2v_cbow_gpu_end_to_end_learns():
    """CBOW + NS fast path end to end on GPU."""
    import multiverso_amd as mv
    from multiverso_amd.apps.wordembedding.model import (WordEmbedding,
                                                         WordEmbeddingOption)
    mv.init()
    torch.manual_seed(1)
    opt = WordEmbeddingOption(embedding_size=64, window=1, negative_num=5,
                              cbow=True, init_learning_rate=0.1,
                              total_words=10_000_000, seed=5,
                              max_groups_per_launch=64)
    model = WordEmbedding(opt, [100] * 20)
    words = torch.stack([torch.arange(0, 20, 2).repeat(100),
                         torch.arange(1, 20, 2).repeat(100)],
                        dim=1).view(-1).cuda()
    sids = (torch.arange(words.numel()) // 10).cuda()
    for _ in range(10):
        model.train_block(words, sids)
    inp = model.input_table.get()
    out = model.output_table.get()
    evens = torch.arange(0, 20, 2, device="cuda")
    pos = torch.sigmoid((inp[evens] * out[evens + 1]).sum(1)).mean()
    wrong = torch.sigmoid((inp[evens] * out[evens.roll(1) + 1]).sum(1)).mean()
    torch.cuda.synchronize()
    # margin only: the tables' random-init stream depends on the global
    # table-id counter (position in the test session), so absolute
    # sigmoid levels shift; discrimination must hold regardless
    # (isolated run measured pos 0.95 / wrong 0.64)
    assert float(pos) > 0.75 and float(pos) - float(wrong) > 0.15, \
        (float(pos), float(wrong))
    mv.shutdown()


def test_w2v_hs_gpu_end_to_end_learns():
    """Hierarchical-softmax path (ragged kernel, Huffman labels) end to
    end on GPU: after training, a (center, context) pair's Huffman-path
    probability must beat a mismatched pair's."""
    import multiverso_amd as mv
    from multiverso_amd.apps.wordembedding.model import (WordEmbedding,
                                                         WordEmbeddingOption)
    mv.init()
    torch.manual_seed(2)
    opt = WordEmbeddingOption(embedding_size=64, window=1, negative_num=0,
                              hs=True, init_learning_rate=0.1,
                              total_words=10_000_000, seed=7,
                              max_groups_per_launch=64)
    model = WordEmbedding(opt, [100] * 20)
    words = torch.stack([torch.arange(0, 20, 2).repeat(100),
                         torch.arange(1, 20, 2).repeat(100)],
                        dim=1).view(-1).cuda()
    sids = (torch.arange(words.numel()) // 10).cuda()
    for _ in range(10):
        model.train_block(words, sids)
    inp = model.input_table.get()
    out = model.output_table.get()

    def path_logprob(ctx, center):
        info = model.huffman.labels[center]
        h = inp[ctx]
        lp = 0.0
        for node, code in zip(info.point, info.code):
            p = torch.sigmoid((h * out[node]).sum())
            # label = 1 - code (model.py _build_hs_tensors)
            lp += float(torch.log((1 - code) * p + code * (1 - p) + 1e-9))
        return lp

    good = sum(path_logprob(e, e + 1) for e in range(0, 20, 2)) / 10
    bad = sum(path_logprob(e, ((e + 3) % 20) | 1) for e in range(0, 20, 2)) / 10
    torch.cuda.synchronize()
    assert good > bad + 0.5, (good, bad)
    mv.shutdown()
