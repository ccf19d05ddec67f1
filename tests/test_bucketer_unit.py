"""Single-process unit tests for GradBucketer's partitioning invariants
(the multi-process all-reduce behavior is covered by test_dist_cpu.py).

Uses a 1-rank gloo group: the bucketer requires an initialized process
group but world_size=1 makes every all-reduce an identity, so the
partition/aliasing logic can be checked deterministically.
"""

import os
import tempfile

import pytest
import torch
import torch.distributed as td

from zaremba_amd.models.lstm_lm import Model
from zaremba_amd.parallel.bucketer import GradBucketer


@pytest.fixture()
def pg():
    f = tempfile.NamedTemporaryFile(delete=False)
    f.close()
    td.init_process_group("gloo", init_method=f"file://{f.name}", rank=0,
                          world_size=1)
    yield
    td.destroy_process_group()
    try:
        os.unlink(f.name)
    except FileNotFoundError:
        pass  # gloo's file-store may remove it on teardown


def test_every_param_in_exactly_one_bucket(pg):
    model = Model(31, 16, 2, dropout=0.0, winit=0.1)
    dp = GradBucketer(model, bucket_bytes=2048)
    seen = set()
    for b in dp.buckets:
        for p in b.params:
            assert id(p) not in seen, "param in two buckets"
            seen.add(id(p))
    assert seen == {id(p) for p in model.parameters() if p.requires_grad}
    dp.detach_hooks()


def test_buckets_follow_reverse_param_order(pg):
    """Buckets are cut along REVERSE parameter order (fc -> layers ->
    embed), the order backward produces grads, so early buckets can
    all-reduce while earlier layers still run backward."""
    model = Model(31, 16, 2, dropout=0.0, winit=0.1)
    dp = GradBucketer(model, bucket_bytes=1)  # one param per bucket
    order = [p for b in dp.buckets for p in b.params]
    expected = list(model.parameters())[::-1]
    assert [id(p) for p in order] == [id(p) for p in expected]
    dp.detach_hooks()


def test_grad_views_alias_flat_buffer(pg):
    model = Model(31, 16, 1, dropout=0.0, winit=0.1)
    dp = GradBucketer(model, bucket_bytes=1 << 20)
    for b in dp.buckets:
        base = b.flat.data_ptr()
        end = base + b.flat.numel() * 4
        total = 0
        for p in b.params:
            g = p.grad
            assert g is not None and g.shape == p.shape
            assert base <= g.data_ptr() < end, "grad not a flat-buffer view"
            total += p.numel()
        assert total == b.flat.numel()
    dp.detach_hooks()


def test_backward_accumulates_into_buckets_and_finalize(pg):
    torch.manual_seed(0)
    model = Model(31, 16, 1, dropout=0.0, winit=0.1)
    dp = GradBucketer(model, bucket_bytes=4096)
    x = torch.randint(0, 31, (4, 3))
    y = torch.randint(0, 31, (4, 3))
    from zaremba_amd import trainer

    dp.zero_grad()
    states = model.state_init(3)
    scores, states = model(x, states)
    trainer.nll_loss(scores, y).backward()
    dp.finalize_backward()  # world=1: identity reduce, must not hang
    for b in dp.buckets:
        assert b.pending == 0, "post-accumulate hooks did not all fire"
    assert any(p.grad.abs().sum() > 0 for p in model.parameters())
    dp.detach_hooks()
