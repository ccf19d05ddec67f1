"""Property-based tests (hypothesis) for the batcher semantics — the
reference's main.py:61-74 contract, checked as stream-level properties
rather than re-implementation: every emitted (x, y) window must be a
contiguous slice of the row-major reshaped stream with y the one-token
shift, windows tile the stream without gaps, and the tail rule (the last
window, full or partial, is dropped) bounds the window count."""

import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from zaremba_amd.data import minibatch, shard_stream


@settings(max_examples=200, deadline=None)
@given(
    n=st.integers(min_value=0, max_value=400),
    B=st.integers(min_value=1, max_value=7),
    T=st.integers(min_value=1, max_value=9),
)
def test_minibatch_properties(n, B, T):
    stream = np.arange(n, dtype=np.int64).reshape(-1, 1)  # distinct tokens
    batches = minibatch(stream, B, T)
    rows = n // B
    grid = torch.arange(rows * B, dtype=torch.int64).view(B, rows)

    pos = 0
    for x, y in batches:
        w = x.size(0)
        assert x.shape == y.shape == (w, B)
        assert 1 <= w <= T
        # x/y are the [pos, pos+w) columns of the stream grid, y shifted
        assert torch.equal(x, grid[:, pos:pos + w].T)
        assert torch.equal(y, grid[:, pos + 1:pos + w + 1].T)
        pos += w

    # tail rule: windows stop strictly before the last column (the final
    # window is dropped), so every emitted token index < rows - 1 and at
    # least one column stays unconsumed
    if batches:
        assert pos < rows - 1 or (pos == rows - 1 and rows >= 1)
        assert all(x.size(0) == T for x, _ in batches[:-1])
    # emitting anything requires at least one full window PLUS a spare
    if rows - 1 < T + 1:
        assert batches == [] or batches[-1][0].size(0) < T

    # EXACT window-count oracle, transcribed from the reference loop
    # (main.py:68-73): window at i iff min(T, rows-1-i) < rows-1-i,
    # i.e. i + T < rows - 1 (deep-fuzzed at 5000 examples in-round)
    expected = sum(1 for i in range(0, max(rows - 1, 0), T)
                   if i + T < rows - 1)
    assert len(batches) == expected, (n, B, T, len(batches), expected)


@settings(max_examples=100, deadline=None)
@given(
    n=st.integers(min_value=0, max_value=300),
    world=st.integers(min_value=1, max_value=8),
)
def test_shard_stream_partitions(n, world):
    stream = np.arange(n, dtype=np.int64).reshape(-1, 1)
    shards = [shard_stream(stream, r, world) for r in range(world)]
    # disjoint, contiguous, order-preserving cover of a stream prefix
    joined = np.concatenate([s.reshape(-1) for s in shards]) if world else []
    assert len(joined) <= n
    assert np.array_equal(joined, np.arange(len(joined)))
    # balanced: sizes differ by at most 1... (equal-size contract)
    sizes = {s.size for s in shards}
    assert len(sizes) <= 2
