"""Distributed data-parallel logic on CPU (gloo, world_size=2).

Verifies the bucketer's all-reduce correctness: two ranks with identical
replicas and different minibatches must end up with identical summed
grads, equal to the single-process sum of the two per-batch grads — the
1-GPU vs N-GPU gradient-equality contract (SURVEY.md §4 'Distributed').
"""

import os
import tempfile

import pytest
import torch
import torch.distributed as td
import torch.multiprocessing as mp

from zaremba_amd.models.lstm_lm import Model
from zaremba_amd.ops import functional as F_ref

V, H, L, B, T = 23, 8, 2, 3, 4


def _make_batch(seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randint(0, V, (T, B), generator=g)
    y = torch.randint(0, V, (T, B), generator=g)
    return x, y


def _rank_main(rank, world, init_file, q):
    td.init_process_group("gloo", init_method=f"file://{init_file}",
                          rank=rank, world_size=world)
    torch.manual_seed(7)
    model = Model(V, H, L, dropout=0.0, winit=0.1)
    from zaremba_amd.parallel.bucketer import GradBucketer
    dp = GradBucketer(model, bucket_bytes=4096)  # force several buckets
    x, y = _make_batch(100 + rank)
    states = model.state_init(B)
    dp.zero_grad()
    scores, _ = model(x, states)
    loss = F_ref.nll_loss(scores, y)
    loss.backward()
    dp.finalize_backward()
    # ship by value (numpy) — CUDA/file-descriptor tensor sharing is not
    # reliable across short-lived spawn workers
    grads = {n: p.grad.numpy().copy() for n, p in model.named_parameters()}
    q.put((rank, grads))
    td.destroy_process_group()


@pytest.mark.timeout(180)
@pytest.mark.parametrize("world", [2, 4])
def test_bucketer_allreduce_matches_serial_sum(world):
    with tempfile.TemporaryDirectory() as d:
        init_file = os.path.join(d, "pg")
        ctx = mp.get_context("spawn")
        q = ctx.SimpleQueue()
        procs = [ctx.Process(target=_rank_main, args=(r, world, init_file, q))
                 for r in range(world)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(world):
            rank, grads = q.get()
            results[rank] = grads
        for p in procs:
            p.join(60)
            assert p.exitcode == 0

    # serial reference: same init, sum of each rank-batch's grads
    torch.manual_seed(7)
    model = Model(V, H, L, dropout=0.0, winit=0.1)
    expected = {n: torch.zeros_like(p) for n, p in model.named_parameters()}
    for seed in range(100, 100 + world):
        model.zero_grad()
        x, y = _make_batch(seed)
        scores, _ = model(x, model.state_init(B))
        F_ref.nll_loss(scores, y).backward()
        for n, p in model.named_parameters():
            expected[n] += p.grad
    for n in expected:
        r0 = torch.from_numpy(results[0][n])
        assert torch.allclose(r0, expected[n], atol=1e-5), n
        for r in range(1, world):
            assert torch.allclose(
                r0, torch.from_numpy(results[r][n]), atol=1e-7), n


# ---------------------------------------------------------------------------
# Distributed ensemble averaging (K13 / BASELINE config 5)
# ---------------------------------------------------------------------------

K_MODELS = 3


def _build_member(num):
    torch.manual_seed(1000 * num)
    return Model(V, H, 1, dropout=0.0, winit=0.1)


def _eval_data(n_batches=3):
    out = []
    for s in range(n_batches):
        out.append(_make_batch(500 + s))
    return out


def _ens_rank_main(rank, world, init_file, q):
    td.init_process_group("gloo", init_method=f"file://{init_file}",
                          rank=rank, world_size=world)
    from zaremba_amd.ensemble_eval import ensemble_perplexity_distributed
    # member i (1-based) lives on rank (i-1) % world — rank 1 owns none
    # when K_MODELS < world leaves a gap; the zero-contribution path is
    # exercised by the k=1 increment (only rank 0 owns model 1).
    my = {f"model {i + 1}": _build_member(i + 1)
          for i in range(K_MODELS) if i % world == rank}
    data = _eval_data()
    ppls = []
    for k in range(1, K_MODELS + 1):
        subset = {name: m for name, m in my.items()
                  if int(name.split()[1]) <= k}
        ppls.append(ensemble_perplexity_distributed(
            data, subset, k, B, V, torch.device("cpu")))
    q.put((rank, ppls))
    td.destroy_process_group()


@pytest.mark.timeout(180)
def test_distributed_ensemble_eval_matches_sequential():
    """Per-rank prob sums + all-reduce(SUM)/k == the sequential
    all-models-on-one-process averaging (reference ensemble.py:97-126),
    for every incremental ensemble size k."""
    world = 2
    with tempfile.TemporaryDirectory() as d:
        init_file = os.path.join(d, "pg")
        ctx = mp.get_context("spawn")
        q = ctx.SimpleQueue()
        procs = [ctx.Process(target=_ens_rank_main,
                             args=(r, world, init_file, q))
                 for r in range(world)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(world):
            rank, ppls = q.get()
            results[rank] = ppls
        for p in procs:
            p.join(60)
            assert p.exitcode == 0

    from zaremba_amd.ensemble_eval import ensemble_perplexity
    data = _eval_data()
    for k in range(1, K_MODELS + 1):
        models = {f"model {i + 1}": _build_member(i + 1) for i in range(k)}
        expected = ensemble_perplexity(data, models, B)
        assert results[0][k - 1] == pytest.approx(expected, rel=1e-6), k
        assert results[1][k - 1] == pytest.approx(expected, rel=1e-6), k
