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


@given(total=st.integers(min_value=16, max_value=100_000),
       n=st.integers(min_value=1, max_value=16),
       seed=st.integers(min_value=0, max_value=1_000))
@settings(max_examples=60, deadline=None)
def test_keyed_partition_matches_shard_owner(total, n, seed):
    """The async engine's keyed plan and the collective plane's owner
    math must agree with ShardSpec.range_of for EVERY id: the id lands
    on the server whose [offset, offset+count) range contains it.
    Domain: total >= n — the tables CHECK size >= num_servers at
    construction (reference array_table.cpp:14), and below that the
    last-server-takes-all layout and the floor-divide owner disagree."""
    import torch
    from multiverso_amd.comm import ShardSpec
    spec = ShardSpec(total, n)
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, total, (min(total, 64),), generator=g)
    base = max(total // n, 1)
    owners = torch.div(ids, base, rounding_mode="floor").clamp_(max=n - 1)
    for i, o in zip(ids.tolist(), owners.tolist()):
        off, cnt = spec.range_of(o)
        assert off <= i < off + cnt, (i, o, off, cnt)
        assert spec.owner_of(i) == o


@given(sizes=st.lists(st.integers(min_value=0, max_value=1 << 20),
                      min_size=1, max_size=8),
       flags=st.data())
@settings(max_examples=30, deadline=None)
def test_addoption_wire_roundtrip(sizes, flags):
    """AddOption's 20-byte envelope and the engine's f64[5] transport
    encoding both round-trip exactly."""
    from multiverso_amd.updaters import AddOption
    from multiverso_amd.async_ps import _opt_from, _opt_tensor
    o = AddOption(worker_id=sizes[0] % 1024, momentum=0.25,
                  learning_rate=0.125, rho=0.5, lambda_=0.0625)
    o2 = AddOption.from_bytes(o.to_bytes())
    assert (o2.worker_id, o2.momentum, o2.learning_rate, o2.rho,
            o2.lambda_) == (o.worker_id, o.momentum, o.learning_rate,
                            o.rho, o.lambda_)
    o3 = _opt_from(_opt_tensor(o))
    assert (o3.worker_id, o3.momentum, o3.learning_rate, o3.rho,
            o3.lambda_) == (o.worker_id, o.momentum, o.learning_rate,
                            o.rho, o.lambda_)
