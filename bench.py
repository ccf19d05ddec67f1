#!/usr/bin/env python3
"""Flagship training benchmark — the driver contract.

Measures training tokens/sec of the Large Zaremba LSTM (2 x 1500, the
BASELINE.json headline config: bs 20/GPU, seq 35, dropout 0.65) on
synthetic vocab-10k data with random-init weights (the PTB train blob is
absent upstream). bf16 compute on the HIP engine, fp32
master weights; full training steps (forward + loss + backward + grad
clip + SGD) — nothing is skipped inside the timed region.

Single GPU:   python bench.py --gpus 1 --steps K --warmup W
Multi-GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
(one rank per GPU over RCCL/xGMI; weak scaling — per-GPU batch fixed at 20).

Rank 0 prints exactly one JSON line with the whole-job aggregate.
"""

import argparse
import json
import os
import time

import numpy as np
import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--hidden_size", type=int, default=1500)
    p.add_argument("--layer_num", type=int, default=2)
    p.add_argument("--batch_size", type=int, default=20)
    p.add_argument("--seq_length", type=int, default=35)
    p.add_argument("--dropout", type=float, default=0.65)
    p.add_argument("--vocab", type=int, default=10000)
    p.add_argument("--lr", type=float, default=1.0)
    p.add_argument("--max_grad_norm", type=float, default=10.0)
    p.add_argument("--engine", type=str, default="auto",
                   choices=["auto", "hip", "eager"])
    p.add_argument("--seed", type=int, default=1234)
    return p.parse_args()


def main():
    args = parse_args()
    from zaremba_amd import data as zdata
    from zaremba_amd import trainer
    from zaremba_amd.models.lstm_lm import Model
    from zaremba_amd.parallel import dist as zdist

    if args.engine == "eager":
        os.environ["ZAREMBA_AMD_FORCE_EAGER"] = "1"
    zdist.maybe_init()
    rank, world = zdist.rank(), zdist.world_size()
    if args.gpus != world:
        # --gpus is authoritative: a mismatch means the launch was wrong
        # (e.g. `bench.py --gpus 8` run single-process would silently
        # bench dp1). Fail loudly instead of reporting the wrong config.
        raise SystemExit(
            f"bench.py: --gpus {args.gpus} but torch.distributed world size "
            f"is {world}. Launch with `python -m torch.distributed.run "
            f"--nnodes=1 --nproc-per-node {args.gpus} --master-addr "
            f"127.0.0.1 bench.py --gpus {args.gpus} ...` (one rank per GPU).")
    has_gpu = torch.cuda.is_available()
    device = torch.device("cuda", zdist.local_rank()) if has_gpu else \
        torch.device("cpu")
    if has_gpu:
        torch.cuda.set_device(device)
    torch.manual_seed(args.seed + rank)

    # synthetic stream, per-rank independent (weak scaling: fixed work/GPU)
    steps_needed = args.steps + args.warmup + 2
    tokens_needed = (steps_needed + 2) * args.batch_size * args.seq_length \
        + args.batch_size * args.seq_length
    rng = np.random.default_rng(args.seed + rank)
    stream = rng.integers(0, args.vocab, size=(tokens_needed * 2, 1),
                          dtype=np.int64)
    batches = zdata.minibatch(stream, args.batch_size, args.seq_length)
    assert len(batches) >= steps_needed, (len(batches), steps_needed)
    batches = [(x.to(device), y.to(device)) for x, y in
               batches[:steps_needed]]

    model = Model(args.vocab, args.hidden_size, args.layer_num, args.dropout,
                  winit=0.04, lstm_type="custom", engine=args.engine)
    model.to(device)
    model.train()

    dp = None
    if world > 1:
        zdist.broadcast_parameters(model)
        from zaremba_amd.parallel.bucketer import GradBucketer
        dp = GradBucketer(model)

    states = model.state_init(args.batch_size)

    def one_step(x, y):
        nonlocal states
        # under DP the bucketer zeroes its flat buffers (grads are views)
        if dp is None:
            model.zero_grad(set_to_none=True)
        else:
            dp.zero_grad()
        states = model.detach(states)
        scores, states = model(x, states)
        loss = trainer.nll_loss(scores, y)
        loss.backward()
        if dp is not None:
            dp.finalize_backward()
        trainer.sgd_step(model, args.lr, args.max_grad_norm)

    bi = 0
    for _ in range(args.warmup):
        x, y = batches[bi % len(batches)]
        one_step(x, y)
        bi += 1

    zdist.barrier()
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        x, y = batches[bi % len(batches)]
        one_step(x, y)
        bi += 1
    if has_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    zdist.barrier()

    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device if has_gpu else "cpu",
                         dtype=torch.float64)
        import torch.distributed as td
        td.all_reduce(t, op=td.ReduceOp.MAX)
        elapsed = t.item()

    tokens = args.steps * args.batch_size * args.seq_length * world
    if rank == 0:
        result = {
            "metric": "train_tokens_per_sec",
            "value": tokens / elapsed,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            # HIP engine = bf16 compute / fp32 masters; the eager
            # comparison path and CPU runs compute in fp32
            "dtype": "bf16" if (has_gpu and args.engine != "eager")
                     else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"zaremba-lstm-large-{args.layer_num}x{args.hidden_size}",
                "global_batch": args.batch_size * world,
                "seq_len": args.seq_length,
                "parallelism": f"dp{world}",
                "vocab": args.vocab,
                "dropout": args.dropout,
            },
        }
        print(json.dumps(result))
    zdist.finalize()


if __name__ == "__main__":
    main()
