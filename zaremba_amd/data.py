"""Data pipeline: PTB word-level corpus + synthetic generator + [T,B] batcher.

Reproduces the reference semantics exactly (cited for parity, not copied):

  * tokenization is ``file[1:].split(' ')`` — the leading character is
    dropped and newlines survive inside tokens, acting as the <eos>
    surrogate (reference ``main.py:44-52``).
  * vocab is ``sorted(set(train_tokens))`` -> word->id dict; valid/test are
    assumed closed over the train vocab (reference ``main.py:53-58``).
  * the batcher truncates the stream to ``B*floor(N/B)`` tokens, reshapes to
    ``[B, N/B]`` (each row a contiguous stream slice), slices length-``T``
    windows and transposes to ``[T, B]`` x/y pairs with y shifted by one.
    The final window is dropped even when it is exactly full (the strict
    ``<`` at reference ``main.py:70``) — verified invariant: a
    929,589-token stream at bs=20, seq=35 yields 1327 windows.
"""

from __future__ import annotations

import os
from typing import List, Tuple

import numpy as np
import torch


def read_tokens(path: str) -> List[str]:
    """Read one PTB split with the reference's exact tokenization."""
    with open(path) as f:
        file = f.read()
    return file[1:].split(" ")


def data_init(data_dir: str = "./data"):
    """Load PTB train/valid/test, build the train vocab, encode all splits.

    Returns ``(trn, vld, tst, vocab_size)`` where each split is an
    ``np.ndarray`` of shape ``[N, 1]`` int64 (reference ``main.py:44-59``).
    """
    train_path = os.path.join(data_dir, "ptb.train.txt")
    if not os.path.exists(train_path):
        raise FileNotFoundError(
            f"{train_path} not found. The PTB train split is not "
            "redistributable here (it is a missing blob in the reference "
            "repo too); place your own copy in the data dir, or run with "
            "--data synthetic for throughput/benchmark work.")
    trn = read_tokens(train_path)
    vld = read_tokens(os.path.join(data_dir, "ptb.valid.txt"))
    tst = read_tokens(os.path.join(data_dir, "ptb.test.txt"))
    words = sorted(set(trn))
    word2id = {w: i for i, w in enumerate(words)}
    trn_ids = [word2id[w] for w in trn]
    vld_ids = [word2id[w] for w in vld]
    tst_ids = [word2id[w] for w in tst]
    return (
        np.array(trn_ids).reshape(-1, 1),
        np.array(vld_ids).reshape(-1, 1),
        np.array(tst_ids).reshape(-1, 1),
        len(words),
    )


def synthetic_init(
    vocab_size: int = 10000,
    train_tokens: int = 929589,
    valid_tokens: int = 73760,
    test_tokens: int = 82430,
    seed: int = 1234,
):
    """Synthetic stand-in for PTB (the train blob is absent upstream; see
    BASELINE.md). Token streams are uniform over the vocab; shapes match
    the real corpus so the batcher produces the same step counts."""
    rng = np.random.default_rng(seed)
    trn = rng.integers(0, vocab_size, size=(train_tokens, 1), dtype=np.int64)
    vld = rng.integers(0, vocab_size, size=(valid_tokens, 1), dtype=np.int64)
    tst = rng.integers(0, vocab_size, size=(test_tokens, 1), dtype=np.int64)
    return trn, vld, tst, vocab_size


def synthetic_markov_init(
    vocab_size: int = 10000,
    branch: int = 20,
    train_tokens: int = 929589,
    valid_tokens: int = 73760,
    test_tokens: int = 82430,
    seed: int = 1234,
):
    """A LEARNABLE synthetic corpus: an order-1 Markov chain where every
    token has exactly ``branch`` equiprobable successors. The optimal
    perplexity is ``branch`` — a trained model approaching it end-to-end
    validates the whole bf16 kernel/training stack (the real PTB train
    split is a missing blob upstream)."""
    rng = np.random.default_rng(seed)
    successors = rng.integers(0, vocab_size, size=(vocab_size, branch),
                              dtype=np.int64)

    def walk(n, state):
        out = np.empty(n, dtype=np.int64)
        choices = rng.integers(0, branch, size=n)
        for i in range(n):
            state = successors[state, choices[i]]
            out[i] = state
        return out, state

    state = 0
    trn, state = walk(train_tokens, state)
    vld, state = walk(valid_tokens, state)
    tst, state = walk(test_tokens, state)
    return (trn.reshape(-1, 1), vld.reshape(-1, 1), tst.reshape(-1, 1),
            vocab_size)


def minibatch(
    data: np.ndarray, batch_size: int, seq_length: int
) -> List[Tuple[torch.Tensor, torch.Tensor]]:
    """Slice a token stream into ``[T, B]`` (x, y) windows.

    Exactly the reference's semantics (``main.py:61-74``) including the
    tail-window drop: a window is emitted only when a *further* token
    remains past it (strict ``<``), so the final window — full or partial —
    is always dropped.
    """
    stream = torch.tensor(np.asarray(data).reshape(-1), dtype=torch.int64)
    rows = stream.size(0) // batch_size
    stream = stream[: rows * batch_size].view(batch_size, rows)
    dataset: List[Tuple[torch.Tensor, torch.Tensor]] = []
    limit = stream.size(1) - 1
    for i in range(0, limit, seq_length):
        window = min(seq_length, limit - i)
        if window < limit - i:
            x = stream[:, i : i + window].transpose(1, 0).contiguous()
            y = stream[:, i + 1 : i + window + 1].transpose(1, 0).contiguous()
            dataset.append((x, y))
    return dataset


def shard_stream(data: np.ndarray, rank: int, world_size: int) -> np.ndarray:
    """Disjoint contiguous shard of a token stream for data parallelism.

    Rank r takes the r-th of ``world_size`` equal contiguous chunks, so
    each DP rank trains on an independent slice (weak scaling: per-rank
    step count shrinks by 1/world_size on real data; synthetic benches
    instead generate a full-size stream per rank).
    """
    flat = np.asarray(data).reshape(-1)
    chunk = len(flat) // world_size
    return flat[rank * chunk : (rank + 1) * chunk].reshape(-1, 1)
