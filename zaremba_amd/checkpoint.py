"""Versioned checkpoint format.

The reference has no serialization at all (its ensemble keeps trained
models only in a dict in memory, ensemble.py:171-176). This defines the
framework's on-disk format: a single ``.pt`` file holding

  * ``format_version``
  * ``state_dict`` keyed by the reference parameter naming
    (``embed.W``, ``rnns.{i}.W_x/W_h/b_x/b_h``, ``fc.W``, ``fc.b``;
    reference model.py:11,26-29,62-63) — fp32 master weights
  * ``model_args`` needed to rebuild the architecture
  * run metadata: epoch, lr, torch RNG state

so ensemble training can parallelize across ranks and any subset of
saved models can be re-loaded for averaged evaluation.
"""

from __future__ import annotations

import torch

FORMAT_VERSION = 1


def save_checkpoint(path: str, model, epoch: int = 0, lr: float = 0.0,
                    extra: dict | None = None):
    payload = {
        "format_version": FORMAT_VERSION,
        "state_dict": {k: v.detach().cpu() for k, v in model.state_dict().items()},
        "model_args": {
            "vocab_size": model.vocab_size,
            "hidden_size": model.hidden_size,
            "layer_num": model.layer_num,
            "dropout": model.dropout_p,
            "winit": model.winit,
            "lstm_type": model.lstm_type,
        },
        "epoch": epoch,
        "lr": lr,
        "torch_rng_state": torch.get_rng_state(),
        # All visible device generators. (The HIP dropout seed derives
        # from the HOST torch generator at HipModel construction, so the
        # torch_rng_state above governs it; these cover eager-path
        # dropout and any torch.cuda sampling.)
        "cuda_rng_state": (torch.cuda.get_rng_state_all()
                           if torch.cuda.is_available() else None),
    }
    if extra:
        payload.update(extra)
    torch.save(payload, path)


def load_checkpoint(path: str, map_location="cpu"):
    payload = torch.load(path, map_location=map_location, weights_only=False)
    if payload.get("format_version") != FORMAT_VERSION:
        raise ValueError(
            f"checkpoint {path}: unsupported format_version "
            f"{payload.get('format_version')!r}")
    return payload


def restore_rng(payload: dict):
    """Restore the RNG streams a checkpoint captured, so a resumed run's
    dropout masks / data draws continue the unbroken run's sequence
    (round-1 gap: the state was saved but never restored)."""
    state = payload.get("torch_rng_state")
    if state is not None:
        torch.set_rng_state(state.cpu() if torch.is_tensor(state) else
                            torch.as_tensor(state, dtype=torch.uint8))
    cuda_state = payload.get("cuda_rng_state")
    if cuda_state is not None and torch.cuda.is_available():
        # Tolerate resuming on a box with fewer devices than the saver.
        n = min(len(cuda_state), torch.cuda.device_count())
        for i in range(n):
            torch.cuda.set_rng_state(cuda_state[i], i)


def build_model_from_checkpoint(path: str, engine: str = "auto"):
    from .models.lstm_lm import Model

    payload = load_checkpoint(path)
    model = Model(engine=engine, **payload["model_args"])
    model.load_state_dict(payload["state_dict"])
    return model, payload
