"""The word-level LSTM language model (Zaremba et al. 2014 architecture).

Architecture parity with the reference (model.py:75-110):
embed -> dropout -> (LSTM -> dropout) x L -> linear, with non-recurrent
dropout only (a fresh per-element mask at every application), uniform
U(-winit, winit) init of every parameter, and per-layer (h, c) state
carried across truncated-BPTT windows.

MI355X-native specifics:
  * parameters are fp32 masters; when the HIP engine runs in bf16 the
    model keeps bf16 shadow copies that the fused clip+SGD kernel
    rewrites in the same pass as the master update,
  * the forward dispatches per-device: CUDA(HIP) tensors run the custom
    CDNA4 kernel path (fused LSTM sequence, MFMA GEMMs, philox dropout);
    CPU tensors run the eager oracle math in ops/functional.py,
  * both --lstm_type values ("pytorch" and "custom") execute the same
    fused HIP cell — the reference's nn.LSTM fast path and its custom
    cell are one implementation here; parameter naming
    always follows the custom convention (W_x/W_h/b_x/b_h per layer).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import _C
from ..ops import functional as F_ref


class Embed(nn.Module):
    """Plain embedding table; forward is a [T,B] -> [T,B,H] gather
    (reference model.py:6-17)."""

    def __init__(self, vocab_size: int, embed_size: int):
        super().__init__()
        self.vocab_size = vocab_size
        self.embed_size = embed_size
        self.W = nn.Parameter(torch.Tensor(vocab_size, embed_size))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F_ref.embedding(self.W, x)

    def __repr__(self):
        return f"Embedding(vocab: {self.vocab_size}, embedding: {self.embed_size})"


class LSTM(nn.Module):
    """One LSTM layer, cuDNN-style dual-bias parameterization, gate order
    (i, f, o, n) (reference model.py:20-55)."""

    def __init__(self, input_size: int, hidden_size: int, dropout: float = 0.0,
                 winit: float = 0.1):
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        # dropout/winit accepted for ctor parity; unused (as in the reference).
        self.W_x = nn.Parameter(torch.Tensor(4 * hidden_size, input_size))
        self.W_h = nn.Parameter(torch.Tensor(4 * hidden_size, hidden_size))
        self.b_x = nn.Parameter(torch.Tensor(4 * hidden_size))
        self.b_h = nn.Parameter(torch.Tensor(4 * hidden_size))

    def forward(self, x, states):
        h0, c0 = states
        out, h, c = F_ref.lstm_layer(x, h0, c0, self.W_x, self.W_h, self.b_x, self.b_h)
        return out, (h, c)

    def __repr__(self):
        return f"LSTM(input: {self.input_size}, hidden: {self.hidden_size})"


class Linear(nn.Module):
    """Output projection; returns 2-D [T*B, V] scores (reference model.py:57-71)."""

    def __init__(self, input_size: int, output_size: int):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.W = nn.Parameter(torch.Tensor(output_size, input_size))
        self.b = nn.Parameter(torch.Tensor(output_size))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F_ref.linear(x, self.W, self.b)

    def __repr__(self):
        return f"FC(input: {self.input_size}, output: {self.output_size})"


class Model(nn.Module):
    def __init__(self, vocab_size: int, hidden_size: int, layer_num: int,
                 dropout: float, winit: float, lstm_type: str = "custom",
                 engine: str = "auto"):
        super().__init__()
        if lstm_type not in ("pytorch", "custom"):
            raise ValueError(f"unknown lstm_type {lstm_type!r}")
        if engine not in ("auto", "hip", "eager"):
            raise ValueError(f"unknown engine {engine!r}")
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.layer_num = layer_num
        self.winit = winit
        self.lstm_type = lstm_type
        self.engine = engine
        self.dropout_p = dropout
        self.embed = Embed(vocab_size, hidden_size)
        self.rnns = nn.ModuleList(
            [LSTM(hidden_size, hidden_size) for _ in range(layer_num)]
        )
        self.fc = Linear(hidden_size, vocab_size)
        self.dropout = nn.Dropout(p=dropout)
        self.reset_parameters()
        self._hip_model = None  # lazy per-device HIP execution plan

    # -- init / state management (reference model.py:90-101) ---------------

    def reset_parameters(self):
        for param in self.parameters():
            nn.init.uniform_(param, -self.winit, self.winit)

    def state_init(self, batch_size: int):
        dev = next(self.parameters()).device
        return [
            (
                torch.zeros(batch_size, layer.hidden_size, device=dev),
                torch.zeros(batch_size, layer.hidden_size, device=dev),
            )
            for layer in self.rnns
        ]

    def detach(self, states):
        return [(h.detach(), c.detach()) for (h, c) in states]

    # -- engine dispatch ----------------------------------------------------

    def _resolve_engine(self, device: torch.device) -> str:
        if self.engine == "eager":
            return "eager"
        if device.type == "cuda":
            if self.engine == "auto" and _C.force_eager_env():
                return "eager"
            return "hip"  # ops raise loudly if the extension is missing
        return "eager"

    def hip(self):
        """Return (building if needed) the HIP execution plan for this model."""
        dev = next(self.parameters()).device
        if self._hip_model is not None and self._hip_model.device != dev:
            self._hip_model = None  # model moved devices: rebuild the plan
        if self._hip_model is None:
            from ..ops.hip_model import HipModel

            self._hip_model = HipModel(self)
        return self._hip_model

    def load_state_dict(self, *a, **kw):
        out = super().load_state_dict(*a, **kw)
        if self._hip_model is not None:
            self._hip_model.invalidate_shadows()
        return out

    # -- forward (reference model.py:103-110) --------------------------------

    def forward(self, x: torch.Tensor, states):
        dev = next(self.parameters()).device
        if x.device != dev:
            x = x.to(dev, non_blocking=True)
        if self._resolve_engine(dev) == "hip":
            return self.hip().forward(x, states, training=self.training)
        return self._forward_eager(x, states)

    def _forward_eager(self, x, states):
        x = self.embed(x)
        x = self.dropout(x)
        for i, rnn in enumerate(self.rnns):
            x, states[i] = rnn(x, states[i])
            x = self.dropout(x)
        scores = self.fc(x)
        return scores, states
