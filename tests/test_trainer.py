"""Trainer-driver semantics: LR schedule off-by-one, training convergence
on a tiny synthetic task, perplexity evaluation, checkpoint round-trip."""

import numpy as np
import pytest
import torch

from zaremba_amd import data as zdata
from zaremba_amd import trainer
from zaremba_amd.checkpoint import build_model_from_checkpoint, save_checkpoint
from zaremba_amd.models.lstm_lm import Model


def _tiny_data(vocab=20, n=600, bs=4, seq=5, seed=0):
    rng = np.random.default_rng(seed)
    # deterministic cyclic stream -> learnable
    stream = (np.arange(n) % vocab).reshape(-1, 1)
    ds = zdata.minibatch(stream, bs, seq)
    return ds, vocab


def test_train_loss_decreases(capsys):
    torch.manual_seed(0)
    ds, vocab = _tiny_data()
    model = Model(vocab, 16, 1, dropout=0.0, winit=0.1, lstm_type="custom")
    ppl0 = trainer.perplexity(ds, model, batch_size=4)
    trainer.train((ds, ds, ds), model, epochs=3, epoch_threshold=100, lr=1.0,
                  factor=1.2, max_norm=5.0, batch_size=4)
    ppl1 = trainer.perplexity(ds, model, batch_size=4)
    assert ppl1 < ppl0 * 0.8, (ppl0, ppl1)


def test_lr_schedule_first_decay_epoch(monkeypatch, capsys):
    """lr decays for every epoch index > factor_epoch: with factor_epoch=1
    the first decayed epoch is index 2 (reference main.py:105-106)."""
    torch.manual_seed(0)
    ds, vocab = _tiny_data(n=200)
    model = Model(vocab, 8, 1, dropout=0.0, winit=0.1)
    seen = []
    orig = trainer.sgd_step

    def spy(model_, lr, max_norm, grad_scale=1.0):
        seen.append(lr)
        return orig(model_, lr, max_norm, grad_scale)

    monkeypatch.setattr(trainer, "sgd_step", spy)
    trainer.train((ds, ds, ds), model, epochs=4, epoch_threshold=1, lr=2.0,
                  factor=2.0, max_norm=5.0, batch_size=4)
    per_epoch = sorted(set(seen), reverse=True)
    assert per_epoch == [2.0, 1.0, 0.5]
    # epochs 0,1 at lr=2.0; epoch 2 at 1.0; epoch 3 at 0.5
    steps = len(ds)
    assert seen[:2 * steps] == [2.0] * 2 * steps
    assert seen[2 * steps:3 * steps] == [1.0] * steps


def test_perplexity_of_uniform_model():
    """An untrained near-zero model scores ~vocab perplexity."""
    ds, vocab = _tiny_data(vocab=32, n=400)
    model = Model(vocab, 8, 1, dropout=0.0, winit=1e-4)
    ppl = trainer.perplexity(ds, model, batch_size=4)
    assert abs(ppl - vocab) / vocab < 0.05


def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(1)
    model = Model(30, 8, 2, dropout=0.3, winit=0.1, lstm_type="custom")
    path = str(tmp_path / "ckpt.pt")
    save_checkpoint(path, model, epoch=7, lr=0.25)
    model2, payload = build_model_from_checkpoint(path)
    assert payload["epoch"] == 7 and payload["lr"] == 0.25
    for (n1, p1), (n2, p2) in zip(model.named_parameters(),
                                  model2.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1, p2)
    # forward parity
    x = torch.randint(0, 30, (4, 3))
    s1 = model.state_init(3)
    s2 = model2.state_init(3)
    model.eval(), model2.eval()
    y1, _ = model(x, s1)
    y2, _ = model2(x, s2)
    assert torch.equal(y1, y2)


def test_resume_continues_lr_schedule(tmp_path, capsys):
    ds, vocab = _tiny_data(n=150)
    torch.manual_seed(0)
    model = Model(vocab, 8, 1, dropout=0.0, winit=0.1)
    # train 4 epochs with decay after epoch 1 and save; then resume at
    # epoch 2 must start from lr 1.0 (2.0 / 2.0)
    path = str(tmp_path / "r.pt")
    save_checkpoint(path, model, epoch=2, lr=2.0)
    model2, payload = build_model_from_checkpoint(path)
    seen = []
    orig = trainer.sgd_step

    def spy(model_, lr, max_norm, grad_scale=1.0):
        seen.append(lr)
        return orig(model_, lr, max_norm, grad_scale)

    import zaremba_amd.trainer as tr
    old = tr.sgd_step
    tr.sgd_step = spy
    try:
        tr.train((ds, ds, ds), model2, epochs=3, epoch_threshold=1, lr=2.0,
                 factor=2.0, max_norm=5.0, batch_size=4,
                 start_epoch=payload["epoch"])
    finally:
        tr.sgd_step = old
    assert set(seen) == {1.0}


def test_ensemble_perplexity_beats_worst_member():
    torch.manual_seed(0)
    ds, vocab = _tiny_data(n=400)
    from zaremba_amd.ensemble_eval import ensemble_perplexity
    models = {}
    ppls = []
    for k in range(2):
        m = Model(vocab, 8, 1, dropout=0.0, winit=0.1)
        trainer.train((ds, ds, ds), m, epochs=1, epoch_threshold=100, lr=0.5,
                      factor=1.2, max_norm=5.0, batch_size=4)
        models[f"model {k + 1}"] = m
        ppls.append(trainer.perplexity(ds, m, batch_size=4))
    ens = ensemble_perplexity(ds, models, batch_size=4)
    assert ens <= max(ppls) + 1e-6


def test_checkpoint_rejects_wrong_version(tmp_path):
    torch.manual_seed(3)
    ds, vocab = _tiny_data()
    model = Model(vocab, 8, 1, dropout=0.0, winit=0.1)
    path = str(tmp_path / "ck.pt")
    save_checkpoint(path, model, epoch=1, lr=0.5)
    payload = torch.load(path, weights_only=False)
    payload["format_version"] = 999
    torch.save(payload, path)
    with pytest.raises(ValueError):
        build_model_from_checkpoint(path)


def test_jsonl_mirror_content(tmp_path):
    """--jsonl writes one machine-readable record per log event with the
    same quantities the console line carries."""
    import json

    torch.manual_seed(4)
    ds, vocab = _tiny_data()
    model = Model(vocab, 8, 1, dropout=0.0, winit=0.1)
    path = str(tmp_path / "log.jsonl")
    trainer.train((ds, ds, ds), model, epochs=3, epoch_threshold=1, lr=1.0,
                  factor=2.0, max_norm=5.0, batch_size=4, jsonl_path=path)
    events = [json.loads(l) for l in open(path)]
    kinds = [e["event"] for e in events]
    assert "step" in kinds and "epoch" in kinds and "final" in kinds
    step = next(e for e in events if e["event"] == "step")
    for f in ("train_loss", "wps", "grad_norm", "lr", "elapsed_s"):
        assert f in step, f
    epochs = [e for e in events if e["event"] == "epoch"]
    assert len(epochs) == 3 and all("valid_ppl" in e for e in epochs)
    # LR decay off-by-one (reference main.py:105-106): decay only for
    # epoch index > factor_epoch, so epochs 0,1 run at lr and epoch 2
    # at lr/factor
    assert epochs[0]["lr"] == epochs[1]["lr"] == pytest.approx(1.0)
    assert epochs[2]["lr"] == pytest.approx(0.5)
    final = next(e for e in events if e["event"] == "final")
    assert "test_ppl" in final


def test_resume_restores_rng_state(tmp_path):
    """restore_rng puts the torch RNG back where the checkpoint left it,
    so a resumed run's dropout/data draws continue the unbroken run's
    sequence (round-1 gap: saved but never restored)."""
    from zaremba_amd.checkpoint import load_checkpoint, restore_rng

    torch.manual_seed(5)
    model = Model(10, 8, 1, dropout=0.0, winit=0.1)
    torch.rand(100)  # advance the stream to a nontrivial point
    expected_next = None
    path = str(tmp_path / "rng.pt")
    state_at_save = torch.get_rng_state().clone()
    save_checkpoint(path, model, epoch=1, lr=1.0)
    expected_next = torch.rand(8)
    # perturb the stream, then restore from the checkpoint
    torch.manual_seed(999)
    torch.rand(13)
    payload = load_checkpoint(path)
    restore_rng(payload)
    assert torch.equal(torch.get_rng_state(), state_at_save)
    assert torch.equal(torch.rand(8), expected_next)


def test_model_ctor_validation():
    with pytest.raises(ValueError):
        Model(10, 8, 1, dropout=0.0, winit=0.1, lstm_type="nonsense")
    with pytest.raises(ValueError):
        Model(10, 8, 1, dropout=0.0, winit=0.1, engine="nonsense")
