"""Training / evaluation drivers.

Reproduces the reference driver semantics exactly (main.py:97-133):
  * per-epoch: fresh zero states, model.train(), LR decay lr/=factor for
    every epoch index > factor_epoch,
  * per-step: zero_grad -> detach(states) -> forward -> nll_loss ->
    backward -> global grad-norm clip (max_norm) -> manual SGD
    ``param -= lr * grad``,
  * logging 10x per epoch with the reference's exact console line
    (train loss normalized by B, wps, grad norm, lr, elapsed minutes,
    peak device memory), per-epoch validation perplexity, final test
    perplexity,
and adds (additively): rank-0-only logging under DP, a machine-readable
JSONL mirror, per-epoch checkpointing with --save/--resume, and a fused
HIP clip+SGD path with fp32 master weights + bf16 shadow rewrite.
"""

from __future__ import annotations

import json
import timeit
from typing import List, Optional, Tuple

import numpy as np
import torch

from .ops import functional as F_ref
from .utils.profiling import trace_range


def nll_loss(scores: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """Stable batch_size-scaled NLL (reference main.py:77-84 semantics).

    Dispatch: fused HIP log-softmax+NLL kernel on GPU scores, eager
    torch math otherwise (CPU, or ZAREMBA_AMD_FORCE_EAGER=1)."""
    from . import _C
    if scores.is_cuda and not _C.force_eager_env():
        from .ops.hip_ops import nll_loss_hip
        return nll_loss_hip(scores, y, y.size(1))
    return F_ref.nll_loss(scores, y)


def perplexity(data, model, batch_size: int) -> float:
    """Evaluate perplexity over a batched split (reference main.py:86-95):
    fresh zero states, state carried across batches, exp(mean(loss/B)).

    Under DP every rank evaluates the full split (replicated, not
    sharded): the carried hidden state makes batch b's input state depend
    on batch b-1, so sharding batches across ranks would change the
    numbers vs the reference. Eval is a per-epoch cost (~1/13 of an
    epoch's batches); replication keeps it semantics-exact."""
    was_training = model.training
    model.eval()
    with torch.no_grad():
        losses = []
        states = model.state_init(batch_size)
        for x, y in data:
            scores, states = model(x, states)
            loss = nll_loss(scores, y.to(scores.device))
            losses.append(loss.item() / batch_size)
    if was_training:
        model.train()
    return float(np.exp(np.mean(losses)))


class _JsonlLogger:
    def __init__(self, path: Optional[str]):
        self.f = open(path, "a") if path else None

    def log(self, record: dict):
        if self.f is not None:
            self.f.write(json.dumps(record) + "\n")
            self.f.flush()

    def close(self):
        if self.f is not None:
            self.f.close()


def _device_mem_gb() -> float:
    if torch.cuda.is_available():
        return torch.cuda.max_memory_allocated() / 1024 / 1024 / 1024
    return 0.0


def sgd_step(model, lr: float, max_norm: float, grad_scale: float = 1.0):
    """Grad clip + manual SGD dispatched per engine.

    Eager: torch math (functional.clip_grad_and_sgd_). HIP: the fused
    multi-tensor norm+update kernels, which also rewrite the bf16 shadow
    weights. Returns the pre-clip grad norm as a 0-d tensor (device-side
    on HIP: only log steps force a sync).
    """
    dev = next(model.parameters()).device
    if dev.type == "cuda" and model._resolve_engine(dev) == "hip":
        return model.hip().clip_and_sgd(lr, max_norm, grad_scale)
    params = [p for p in model.parameters() if p.grad is not None]
    if grad_scale != 1.0:
        for p in params:
            p.grad.mul_(grad_scale)
    return F_ref.clip_grad_and_sgd_(list(model.parameters()), max_norm, lr)


def train(
    data: Tuple[List, List, List],
    model,
    epochs: int,
    epoch_threshold: int,
    lr: float,
    factor: float,
    max_norm: float,
    batch_size: int,
    log_every: Optional[int] = None,
    model_num: Optional[int] = None,
    jsonl_path: Optional[str] = None,
    save_path: Optional[str] = None,
    start_epoch: int = 0,
    dp=None,
    is_rank0: bool = True,
):
    """The training loop (reference main.py:97-133 / ensemble.py:128-164).

    ``log_every=None`` -> the reference's len(trn)//10 cadence; ensemble
    mode passes 800. ``dp`` is an optional parallel.Bucketer handling
    gradient all-reduce overlap; ``start_epoch``/``save_path`` implement
    resume (additive; the reference has no checkpointing).
    """
    trn, vld, tst = data
    tic = timeit.default_timer()
    total_words = 0
    jlog = _JsonlLogger(jsonl_path if is_rank0 else None)
    banner = "Starting training.\n" if model_num is None else (
        f"Starting training of model {model_num}.\n")
    if is_rank0:
        print(banner)
    # Reconstruct the LR the resumed epoch should start from.
    for epoch in range(start_epoch):
        if epoch > epoch_threshold:
            lr = lr / factor
    cadence = log_every if log_every is not None else max(1, len(trn) // 10)
    for epoch in range(start_epoch, epochs):
        states = model.state_init(batch_size)
        model.train()
        if epoch > epoch_threshold:
            lr = lr / factor
        for i, (x, y) in enumerate(trn):
            total_words += x.numel()
            # set_to_none saves a fill + an accumulate add per param; under
            # DP the bucketer zeroes its flat buffers instead (one fill per
            # bucket, grads are views into them).
            if dp is None:
                model.zero_grad(set_to_none=True)
            else:
                dp.zero_grad()
            states = model.detach(states)
            with trace_range("forward"):
                scores, states = model(x, states)
            with trace_range("loss"):
                loss = nll_loss(scores, y.to(scores.device))
            with trace_range("backward"):
                loss.backward()
            if dp is not None:
                with trace_range("allreduce_join"):
                    dp.finalize_backward()
            # Under DP each rank holds the SUM of per-rank grads after the
            # all-reduce. The reference loss SUMS over the batch dimension
            # (main.py:82-84), so summed grads are exactly the gradients
            # the reference would compute at batch_size = world * B — the
            # "global batch 160" semantics of BASELINE config 4. No
            # averaging (grad_scale stays 1).
            with trace_range("clip_sgd"):
                norm = sgd_step(model, lr, max_norm)
            if i % cadence == 0:
                if model._hip_model is not None:
                    model._hip_model.check_aborts()
            if i % cadence == 0 and is_rank0:
                toc = timeit.default_timer()
                norm_v = float(norm)
                loss_v = loss.item() / batch_size
                wps = round(total_words / (toc - tic))
                print(
                    "batch no = {:d} / {:d}, ".format(i, len(trn))
                    + "train loss = {:.3f}, ".format(loss_v)
                    + "wps = {:d}, ".format(wps)
                    + "dw.norm() = {:.3f}, ".format(norm_v)
                    + "lr = {:.3f}, ".format(lr)
                    + "since beginning = {:d} mins, ".format(round((toc - tic) / 60))
                    + "cuda memory = {:.3f} GBs".format(_device_mem_gb())
                )
                jlog.log({
                    "event": "step", "epoch": epoch, "batch": i,
                    "train_loss": loss_v, "wps": wps, "grad_norm": norm_v,
                    "lr": lr, "elapsed_s": toc - tic,
                })
        model.eval()
        # Epoch boundary: surface any bounded-spin aborts before the eval
        # result is trusted (log-step checks cover only cadence points).
        if model._hip_model is not None:
            model._hip_model.check_aborts()
        val_perp = perplexity(vld, model, batch_size)
        if is_rank0:
            print("Epoch : {:d} || Validation set perplexity : {:.3f}".format(
                epoch + 1, val_perp))
            print("*************************************************\n")
            jlog.log({"event": "epoch", "epoch": epoch, "valid_ppl": val_perp,
                      "lr": lr})
            if save_path:
                from .checkpoint import save_checkpoint
                save_checkpoint(save_path, model, epoch=epoch + 1, lr=lr)
    tst_perp = perplexity(tst, model, batch_size)
    if model._hip_model is not None:
        model._hip_model.check_aborts()
    if is_rank0:
        print("Test set perplexity : {:.3f}".format(tst_perp))
        print("Training is over." if model_num is None else
              f"Model {model_num} is trained!\n")
        jlog.log({"event": "final", "test_ppl": tst_perp})
    jlog.close()
    return tst_perp
