"""Eager (pure-PyTorch) implementations of every op in the framework.

These are the semantic ground truth: the math transcribes the reference
implementation exactly (citations per function), in a numerically stable
form where the reference is naive. They serve three roles:

  1. the CPU execution path (config 1: plumbing, no GPU),
  2. the fp32 oracle that every HIP kernel is unit-tested against,
  3. the fallback when running on non-gfx950 devices for debugging.

The HIP kernel path lives in zaremba_amd/ops/hip_ops.py + csrc/.
"""

from __future__ import annotations

from typing import List, Tuple

import torch


# ---------------------------------------------------------------------------
# LSTM cell (reference model.py:34-45; gate order i, f, o, n)
# ---------------------------------------------------------------------------

def lstm_step(
    x: torch.Tensor,
    h: torch.Tensor,
    c: torch.Tensor,
    W_x: torch.Tensor,
    W_h: torch.Tensor,
    b_x: torch.Tensor,
    b_h: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """One LSTM timestep, gate order (input, forget, output, new).

    Note this differs from nn.LSTM's (i, f, g, o) order — state_dicts are
    not interchangeable between implementations (reference model.py:37-41).
    """
    gx = torch.addmm(b_x, x, W_x.t())
    gh = torch.addmm(b_h, h, W_h.t())
    xi, xf, xo, xn = gx.chunk(4, 1)
    hi, hf, ho, hn = gh.chunk(4, 1)
    i = torch.sigmoid(xi + hi)
    f = torch.sigmoid(xf + hf)
    o = torch.sigmoid(xo + ho)
    n = torch.tanh(xn + hn)
    c = f * c + i * n
    h = o * torch.tanh(c)
    return h, c


def lstm_layer(
    x: torch.Tensor,
    h0: torch.Tensor,
    c0: torch.Tensor,
    W_x: torch.Tensor,
    W_h: torch.Tensor,
    b_x: torch.Tensor,
    b_h: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Unrolled LSTM layer over a [T, B, X] input (reference model.py:48-55)."""
    h, c = h0, c0
    outputs = []
    for x_t in x.unbind(0):
        h, c = lstm_step(x_t, h, c, W_x, W_h, b_x, b_h)
        outputs.append(h)
    return torch.stack(outputs), h, c


# ---------------------------------------------------------------------------
# Embedding (reference model.py:6-17)
# ---------------------------------------------------------------------------

def embedding(W: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """[T,B] int64 -> [T,B,H] gather from the [V,H] table."""
    return W[x]


# ---------------------------------------------------------------------------
# Output projection (reference model.py:57-71)
# ---------------------------------------------------------------------------

def linear(x: torch.Tensor, W: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Flattens [T,B,H] -> [T*B,H]; returns 2-D [T*B, V] scores."""
    return torch.addmm(b, x.view(-1, x.size(-1)), W.t())


# ---------------------------------------------------------------------------
# Loss (reference main.py:77-84)
# ---------------------------------------------------------------------------

def nll_loss(scores: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """batch_size x cross-entropy: sums over batch, averages over time.

    The reference computes a numerically naive softmax (exp then
    normalize, main.py:79-80) which overflows outside fp32; this is the
    same math through a max-subtracted log-softmax. Scale convention:
    mean over the T*B tokens of (-log p) times B, i.e. the per-timestep
    loss is summed over the batch and averaged over time (main.py:82-84).
    """
    batch_size = y.size(1)
    logp = torch.log_softmax(scores.float(), dim=1)
    answer_logp = logp[torch.arange(y.numel(), device=y.device), y.reshape(-1)]
    return torch.mean(-answer_logp * batch_size)


def ensemble_nll_loss(scores_list: List[torch.Tensor], y: torch.Tensor) -> torch.Tensor:
    """NLL of the arithmetic mean of per-model probabilities
    (reference ensemble.py:97-109), via stable softmax."""
    batch_size = y.size(1)
    probs = torch.stack([torch.softmax(s.float(), dim=1) for s in scores_list])
    mean_probs = probs.mean(dim=0)
    answer = mean_probs[torch.arange(y.numel(), device=y.device), y.reshape(-1)]
    return torch.mean(-torch.log(answer) * batch_size)


# ---------------------------------------------------------------------------
# Optimizer step (reference main.py:115-117)
# ---------------------------------------------------------------------------

def clip_grad_and_sgd_(
    params: List[torch.Tensor], max_norm: float, lr: float
) -> torch.Tensor:
    """Global L2 grad-norm clip followed by vanilla SGD, in place.

    Matches torch.nn.utils.clip_grad_norm_ + the reference's manual
    ``param -= lr * param.grad`` (main.py:115-117). Returns the
    pre-clip total norm (the value the reference logs as dw.norm()).
    """
    grads = [p.grad for p in params if p.grad is not None]
    total_norm = torch.norm(torch.stack([torch.norm(g.detach(), 2.0) for g in grads]), 2.0)
    clip_coef = max_norm / (total_norm + 1e-6)
    coef = torch.clamp(clip_coef, max=1.0)
    with torch.no_grad():
        for p in params:
            if p.grad is not None:
                p.add_(p.grad, alpha=-lr * float(coef))
    return total_norm
