#!/usr/bin/env python3
"""Comparison baseline: the SAME Large training step implemented with
stock PyTorch modules — torch.nn.LSTM (MIOpen fused RNN on ROCm),
nn.Embedding, nn.Dropout, nn.Linear, F.cross_entropy, clip_grad_norm_ +
manual SGD — i.e. the strongest library-only implementation of the
reference's step (reference README.md:29 notes nn.LSTM was ~2x the
custom cell on cuDNN). This is the denominator for the "match or beat"
claim: our hand-written HIP engine vs what a user gets from PyTorch-ROCm
without it, on the same GPU, same shapes, same timing protocol as
bench.py (full step; barrier+sync bracketed; one JSON line).

Usage: python tools/bench_nnlstm.py [--steps 60] [--warmup 10]
       [--dtype bf16|fp32]   (bf16 runs the model in bf16 autocast-free:
       weights/activations cast, fp32 master copy updated by SGD — the
       same master-weight scheme our HIP engine uses)
"""

import argparse
import json
import sys
import time

import torch
import torch.nn as nn
import torch.nn.functional as F


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=60)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--hidden_size", type=int, default=1500)
    p.add_argument("--layer_num", type=int, default=2)
    p.add_argument("--batch_size", type=int, default=20)
    p.add_argument("--seq_length", type=int, default=35)
    p.add_argument("--dropout", type=float, default=0.65)
    p.add_argument("--vocab", type=int, default=10000)
    p.add_argument("--lr", type=float, default=1.0)
    p.add_argument("--max_grad_norm", type=float, default=10.0)
    p.add_argument("--dtype", choices=["bf16", "fp32"], default="bf16")
    return p.parse_args()


class NnLstmLm(nn.Module):
    """embed -> drop -> (nn.LSTM layer -> drop) x L -> fc, matching the
    reference composition (model.py:103-110) with the library fast path."""

    def __init__(self, V, H, L, p):
        super().__init__()
        self.embed = nn.Embedding(V, H)
        self.rnns = nn.ModuleList([nn.LSTM(H, H) for _ in range(L)])
        self.drop = nn.Dropout(p)
        self.fc = nn.Linear(H, V)
        for prm in self.parameters():
            nn.init.uniform_(prm, -0.04, 0.04)

    def forward(self, x, states):
        h = self.drop(self.embed(x))
        new_states = []
        for rnn, s in zip(self.rnns, states):
            h, ns = rnn(h, s)
            new_states.append(ns)
            h = self.drop(h)
        return self.fc(h.reshape(-1, h.size(-1))), new_states


def main():
    args = parse_args()
    if not torch.cuda.is_available():
        print("no GPU", file=sys.stderr)
        sys.exit(1)
    dev = torch.device("cuda:0")
    torch.manual_seed(1234)
    V, H, L = args.vocab, args.hidden_size, args.layer_num
    B, T = args.batch_size, args.seq_length

    model = NnLstmLm(V, H, L, args.dropout).to(dev)
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    masters = None
    if dtype == torch.bfloat16:
        # fp32 masters + bf16 compute weights (same scheme as our engine)
        masters = [p.detach().clone() for p in model.parameters()]
        model.to(torch.bfloat16)

    xs = torch.randint(0, V, (T, B), device=dev)
    ys = torch.randint(0, V, (T * B,), device=dev)
    states = [(torch.zeros(1, B, H, device=dev, dtype=dtype),
               torch.zeros(1, B, H, device=dev, dtype=dtype))
              for _ in range(L)]

    def one_step():
        nonlocal states
        model.zero_grad(set_to_none=True)
        states = [(h.detach(), c.detach()) for h, c in states]
        scores, states = model(xs, states)
        # reference loss semantics: B x mean CE over T*B tokens
        loss = F.cross_entropy(scores.float(), ys) * B
        loss.backward()
        torch.nn.utils.clip_grad_norm_(model.parameters(),
                                       args.max_grad_norm)
        with torch.no_grad():
            if masters is None:
                for p in model.parameters():
                    p.add_(p.grad, alpha=-args.lr)
            else:
                for m, p in zip(masters, model.parameters()):
                    m.add_(p.grad.float(), alpha=-args.lr)
                    p.copy_(m)

    for _ in range(args.warmup):
        one_step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    tokens = args.steps * B * T
    print(json.dumps({
        "metric": "train_tokens_per_sec",
        "value": tokens / elapsed,
        "unit": "tokens/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": elapsed / args.steps * 1000.0,
        "engine": f"torch-nn.LSTM(MIOpen)-{args.dtype}",
        "config": {"model": f"zaremba-lstm-large-{L}x{H}",
                   "global_batch": B, "seq_len": T, "vocab": V,
                   "dropout": args.dropout},
    }))


if __name__ == "__main__":
    main()
