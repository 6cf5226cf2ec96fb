"""Property-style tests (hypothesis) for the sharding/sparse primitives —
the pieces whose index math guards multi-GPU correctness."""
import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from autodist_amd.parallel.comm import coalesce_rows
from autodist_amd.parallel.partitioner import make_shard_slices, \
    split_boundaries


@given(n=st.integers(1, 1000), world=st.integers(1, 9))
@settings(max_examples=80, deadline=None)
def test_split_boundaries_partition(n, world):
    bounds = split_boundaries(n, world)
    # shards clamp to the dimension (never more shards than rows)
    assert len(bounds) == min(world, n)
    assert bounds[0][0] == 0 and bounds[-1][1] == n
    for (s0, e0), (s1, e1) in zip(bounds, bounds[1:]):
        assert e0 == s1 and e0 >= s0 and e1 >= s1
    # balanced: sizes differ by at most 1
    sizes = [e - s for s, e in bounds]
    assert max(sizes) - min(sizes) <= 1


@given(rows=st.integers(2, 40), dim=st.integers(1, 6),
       nnz=st.integers(0, 60), seed=st.integers(0, 2**16))
@settings(max_examples=60, deadline=None)
def test_coalesce_rows_matches_dense_scatter(rows, dim, nnz, seed):
    torch.manual_seed(seed)
    idx = torch.randint(0, rows, (nnz,))
    vals = torch.randn(nnz, dim)
    uniq, summed = coalesce_rows(idx, vals)
    dense = torch.zeros(rows, dim)
    dense.index_add_(0, idx, vals)
    ref_rows = dense.abs().sum(1).nonzero().flatten()
    # every unique touched row appears exactly once with the summed value
    assert sorted(uniq.tolist()) == uniq.tolist()
    got = torch.zeros(rows, dim)
    got[uniq] = summed
    # rows that sum to exactly zero may legitimately appear in uniq
    assert torch.allclose(got[ref_rows], dense[ref_rows], atol=1e-5)


@given(n0=st.integers(1, 64), n1=st.integers(1, 8),
       shards=st.integers(1, 6), axis=st.integers(0, 1),
       seed=st.integers(0, 2**16))
@settings(max_examples=60, deadline=None)
def test_shard_slices_reassemble(n0, n1, shards, axis, seed):
    shape = (n0, n1)
    if shape[axis] < shards:
        shards = shape[axis]
    parts = [1, 1]
    parts[axis] = shards
    spec = ",".join(str(p) for p in parts)
    slices = make_shard_slices(shape, spec)
    torch.manual_seed(seed)
    t = torch.randn(*shape)
    views = [sl.view(t) for sl in slices]
    assert sum(v.shape[axis] for v in views) == shape[axis]
    re = torch.cat(views, dim=axis)
    assert torch.equal(re, t)
