"""Data-pipeline invariants verified against the reference semantics
(SURVEY.md §2.1 components 4-5; reference main.py:44-74)."""

import numpy as np
import pytest
import torch

from zaremba_amd import data as zdata


def test_minibatch_shapes_and_shift():
    stream = np.arange(1000).reshape(-1, 1)
    ds = zdata.minibatch(stream, batch_size=4, seq_length=7)
    assert len(ds) > 0
    for x, y in ds:
        assert x.shape == (7, 4)
        assert y.shape == (7, 4)
        assert x.dtype == torch.int64
    # y is x shifted by one within each row's contiguous stream slice.
    x0, y0 = ds[0]
    assert torch.equal(y0[:-1], x0[1:])


def test_minibatch_rows_are_contiguous_stream_slices():
    stream = np.arange(100).reshape(-1, 1)
    ds = zdata.minibatch(stream, batch_size=2, seq_length=5)
    # rows = 50 per batch row; row 0 covers tokens 0..49, row 1 covers 50..99
    x0, _ = ds[0]
    assert x0[0, 0].item() == 0
    assert x0[0, 1].item() == 50
    assert x0[1, 0].item() == 1


def test_minibatch_tail_window_drop():
    # Verified reference invariant (strict < at main.py:70): a PTB-sized
    # stream of 929,589 tokens at bs=20, seq=35 yields exactly 1327 windows.
    stream = np.zeros((929589, 1), dtype=np.int64)
    ds = zdata.minibatch(stream, batch_size=20, seq_length=35)
    assert len(ds) == 1327


def test_minibatch_drops_final_full_window():
    # stream rows of length 71: limit=70 = 10 full windows of 7, but the
    # final window has window == limit - i and is dropped (strict <).
    stream = np.arange(71).reshape(-1, 1)
    ds = zdata.minibatch(stream, batch_size=1, seq_length=7)
    assert len(ds) == 9


def test_ptb_valid_tokenization():
    # The '\n'-as-token semantics: valid split has exactly 73,760 tokens
    # and at least one token containing a newline (the <eos> surrogate).
    toks = zdata.read_tokens("data/ptb.valid.txt")
    assert len(toks) == 73760
    assert any("\n" in t for t in toks)


def test_synthetic_shapes():
    trn, vld, tst, v = zdata.synthetic_init(vocab_size=100, train_tokens=5000,
                                            valid_tokens=500, test_tokens=600)
    assert v == 100
    assert trn.shape == (5000, 1)
    assert trn.max() < 100 and trn.min() >= 0


def test_shard_stream_disjoint():
    stream = np.arange(100).reshape(-1, 1)
    shards = [zdata.shard_stream(stream, r, 4) for r in range(4)]
    cat = np.concatenate([s.reshape(-1) for s in shards])
    assert len(cat) == 100
    assert len(set(cat.tolist())) == 100


def test_synthetic_markov_learnable_structure():
    trn, vld, tst, v = zdata.synthetic_markov_init(
        vocab_size=200, branch=5, train_tokens=3000, valid_tokens=300,
        test_tokens=300)
    assert v == 200
    # every observed bigram must be one of the 5 successors of its
    # predecessor (the property that makes perplexity=branch achievable)
    flat = trn.reshape(-1)
    import collections
    succ = collections.defaultdict(set)
    for a, b in zip(flat[:-1], flat[1:]):
        succ[int(a)].add(int(b))
    assert max(len(s) for s in succ.values()) <= 5
