#!/usr/bin/env python3
"""Probe: count k_pad GEMM engagement during one Large train step and
time the step phases. Run on a GPU box."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import zaremba_amd.ops.hip_ops as hip_ops
from zaremba_amd.models.lstm_lm import Model
from zaremba_amd import trainer

real_ext = hip_ops.ext
counts = {"pad": 0, "nopad": 0, "splitk": 0}


class Proxy:
    def __init__(self, e):
        self._e = e

    def gemm(self, *a, **kw):
        k_pad = kw.get("k_pad", a[6] if len(a) > 6 else 0)
        counts["pad" if k_pad else "nopad"] += 1
        return self._e.gemm(*a, **kw)

    def gemm_splitk(self, *a, **kw):
        counts["splitk"] += 1
        return self._e.gemm_splitk(*a, **kw)

    def __getattr__(self, n):
        return getattr(self._e, n)


hip_ops.ext = lambda: Proxy(real_ext())

T, B, H, V = 35, 20, 1500, 10000
torch.manual_seed(0)
model = Model(V, H, 2, dropout=0.65, winit=0.04, engine="hip").to("cuda")
x = torch.randint(0, V, (T, B), device="cuda")
y = torch.randint(0, V, (T, B), device="cuda")
states = model.state_init(B)
model.train()

hm = model.hip()
for it in range(13):
    if it == 3:
        counts["pad"] = counts["nopad"] = counts["splitk"] = 0
        torch.cuda.synchronize()
        t0 = time.perf_counter()
    states = model.detach(states)
    scores, states = model(x, states)
    loss = trainer.nll_loss(scores, y)
    for p in model.parameters():
        p.grad = None
    loss.backward()
    hm.clip_and_sgd(lr=1.0, max_norm=10.0)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 10
print(f"{dt*1e3:.3f} ms/step; gemm calls per 10 steps: {counts}")
