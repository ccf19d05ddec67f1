"""Ensemble (model-averaging) evaluation (reference ensemble.py:97-126).

Per batch: forward every model (each with its own carried hidden state),
average the per-model probability tensors arithmetically, take the NLL of
the mean. Perplexity bookkeeping matches perplexity() in trainer.py.
"""

from __future__ import annotations

from typing import Dict

import numpy as np
import torch

from .ops import functional as F_ref


def ensemble_perplexity(data, models: Dict[str, torch.nn.Module],
                        batch_size: int) -> float:
    with torch.no_grad():
        for m in models.values():
            m.eval()
        losses = []
        states = {name: m.state_init(batch_size) for name, m in models.items()}
        for x, y in data:
            scores = []
            for name, m in models.items():
                score, states[name] = m(x, states[name])
                scores.append(score)
            loss = F_ref.ensemble_nll_loss(scores, y.to(scores[0].device))
            losses.append(loss.item() / batch_size)
    return float(np.exp(np.mean(losses)))
