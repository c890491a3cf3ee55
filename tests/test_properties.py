"""Property-based tests (hypothesis) for pure logic: sharding partition
math (reference array_table.cpp:11-21 / matrix_table.cpp:276) and the
SparseFilter wire format."""

import torch
from hypothesis import given, settings, strategies as st

from multiverso_amd.comm import ShardSpec


@given(total=st.integers(min_value=1, max_value=10_000),
       n=st.integers(min_value=1, max_value=64))
@settings(max_examples=200, deadline=None)
def test_shard_spec_partitions_exactly(total, n):
    if total < n:
        return  # tables CHECK total >= n at construction
    spec = ShardSpec(total, n)
    assert sum(spec.counts) == total
    assert spec.offsets[0] == 0
    for r in range(1, n):
        assert spec.offsets[r] == spec.offsets[r - 1] + spec.counts[r - 1]
    # reference: size/n each, remainder on the LAST server
    base = total // n
    assert all(c == base for c in spec.counts[:-1])
    assert spec.counts[-1] == total - base * (n - 1)
    # owner_of agrees with the ranges
    for idx in {i for i in (0, total - 1, total // 2, base,
                            max(0, base - 1)) if i < total}:
        o = spec.owner_of(idx)
        off, cnt = spec.range_of(o)
        assert off <= idx < off + cnt, (idx, o, off, cnt)


@given(n=st.integers(min_value=1, max_value=512),
       density=st.floats(min_value=0.0, max_value=1.0),
       seed=st.integers(min_value=0, max_value=2**31))
@settings(max_examples=100, deadline=None)
def test_sparse_filter_roundtrip_any_density(n, density, seed):
    from multiverso_amd import sparse_filter as sf
    g = torch.Generator().manual_seed(seed)
    v = torch.randn(n, generator=g)
    mask = torch.rand(n, generator=g) < density
    v = v * mask
    payload, comp = sf.filter_in(v)
    out = sf.filter_out(payload, comp, n)
    assert torch.equal(out, v)
    if comp:  # compression only when it actually shrinks the payload
        assert payload.numel() < n


@given(counts=st.lists(st.integers(min_value=1, max_value=10_000),
                       min_size=2, max_size=120),
       )
@settings(max_examples=50, deadline=None)
def test_huffman_invariants(counts):
    from multiverso_amd.apps.wordembedding.huffman import HuffmanEncoder
    enc = HuffmanEncoder()
    enc.build_from_term_frequency(counts)
    v = len(counts)
    codes = ["".join(map(str, l.code)) for l in enc.labels]
    # prefix-free
    sc = sorted(codes)
    for a, b in zip(sc, sc[1:]):
        assert not b.startswith(a)
    # weighted path length is optimal-ish: shorter codes for higher counts
    # (weak: the max-count word's code is no longer than the min-count's)
    import numpy as np
    mx, mn = int(np.argmax(counts)), int(np.argmin(counts))
    assert len(codes[mx]) <= len(codes[mn])
    for l in enc.labels:
        assert len(l.point) == len(l.code) >= 1
        assert all(0 <= p < v - 1 for p in l.point)
        assert l.point[0] == v - 2  # root first (reference layout)
