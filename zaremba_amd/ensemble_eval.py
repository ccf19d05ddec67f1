"""Ensemble (model-averaging) evaluation (reference ensemble.py:97-126).

Per batch: forward every model (each with its own carried hidden state),
average the per-model probability tensors arithmetically, take the NLL of
the mean. Perplexity bookkeeping matches perplexity() in trainer.py.

Two execution shapes:

  * ``ensemble_perplexity`` — single process, all models local. On GPU
    the per-model softmax is accumulated by the fused K13 kernel
    (``softmax_acc``: acc += softmax(scores) in one launch per member —
    no per-model probability tensors, no k-way stack).
  * ``ensemble_perplexity_distributed`` — one process per GPU: every
    rank forwards only the members IT holds, accumulates their
    probability sum locally, and ONE RCCL all-reduce(SUM) per batch over
    the [N, V] partial sums merges the ensemble (then /k). This is
    BASELINE config 5's eval path: all 8 GPUs work on every batch and
    the only traffic is one 28 MB fp32 tensor per batch over xGMI.
"""

from __future__ import annotations

from typing import Dict

import numpy as np
import torch

from . import _C


def _prob_sum(models: Dict[str, torch.nn.Module], states, x, acc):
    """acc += sum over models of softmax(model(x)); returns None.
    Uses the fused HIP kernel on GPU scores, eager softmax otherwise."""
    for name, m in models.items():
        score, states[name] = m(x, states[name])
        if score.is_cuda and not _C.force_eager_env():
            _C.ext().softmax_acc(score.float(), acc)
        else:
            acc += torch.softmax(score.float(), dim=1)


def _mean_prob_nll(acc, k: int, y) -> torch.Tensor:
    """NLL of the k-member mean probabilities (reference ensemble.py:106-109
    math: mean over tokens of -log(mean prob of the answer) * batch_size)."""
    batch_size = y.size(1)
    answer = acc[torch.arange(y.numel(), device=acc.device),
                 y.reshape(-1).to(acc.device)] / k
    return torch.mean(-torch.log(answer) * batch_size)


def ensemble_perplexity(data, models: Dict[str, torch.nn.Module],
                        batch_size: int) -> float:
    with torch.no_grad():
        for m in models.values():
            m.eval()
        losses = []
        states = {name: m.state_init(batch_size) for name, m in models.items()}
        acc = None
        for x, y in data:
            first = next(iter(models.values()))
            dev = next(first.parameters()).device
            N = x.numel()
            if acc is None or acc.size(0) != N:
                acc = torch.zeros(N, first.vocab_size, dtype=torch.float32,
                                  device=dev)
            else:
                acc.zero_()
            _prob_sum(models, states, x, acc)
            loss = _mean_prob_nll(acc, len(models), y)
            losses.append(loss.item() / batch_size)
    return float(np.exp(np.mean(losses)))


def ensemble_perplexity_distributed(data, my_models: Dict[str, torch.nn.Module],
                                    k_total: int, batch_size: int,
                                    vocab_size: int, device) -> float:
    """Distributed ensemble averaging: each rank holds a disjoint subset
    of the k_total members (possibly none); per batch the ranks' partial
    probability sums are merged by all_reduce(SUM) and every rank takes
    the NLL of the mean. Returns the identical perplexity on all ranks.

    Math identity with ensemble_perplexity is pinned by
    tests/test_dist_cpu.py::test_distributed_ensemble_eval_matches_sequential.
    """
    import torch.distributed as td

    if not (td.is_available() and td.is_initialized()):
        raise RuntimeError("ensemble_perplexity_distributed requires an "
                           "initialized process group")
    with torch.no_grad():
        for m in my_models.values():
            m.eval()
        losses = []
        states = {name: m.state_init(batch_size)
                  for name, m in my_models.items()}
        acc = None
        for x, y in data:
            N = x.numel()
            if acc is None or acc.size(0) != N:
                acc = torch.zeros(N, vocab_size, dtype=torch.float32,
                                  device=device)
            else:
                acc.zero_()
            _prob_sum(my_models, states, x, acc)
            td.all_reduce(acc, op=td.ReduceOp.SUM)
            loss = _mean_prob_nll(acc, k_total, y)
            losses.append(loss.item() / batch_size)
    return float(np.exp(np.mean(losses)))
