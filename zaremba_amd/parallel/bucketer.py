"""Bucketed gradient all-reduce overlapped with the BPTT backward.

Design (MI355X/xGMI-first, SURVEY.md §5 'Distributed communication'):
  * parameters are grouped into buckets in REVERSE parameter order — the
    order backward produces grads (fc -> layer L..1 -> embed) — so each
    bucket's all-reduce launches while earlier layers are still running
    their backward kernels,
  * each bucket owns one persistent flat fp32 buffer; ``param.grad`` is a
    view into it, so autograd accumulates straight into the communication
    buffer (no pack/unpack copies),
  * the per-param post-accumulate-grad hook counts the bucket down and
    fires an async RCCL all-reduce (SUM) the moment the bucket completes;
    ``finalize_backward()`` joins all in-flight works before the fused
    clip+SGD consumes the grads. Grads stay SUMMED across ranks — no
    division by world_size anywhere: the reference loss sums over the
    batch dimension, so summed grads are exactly the reference's
    gradients at batch_size = world * B (trainer.py's grad_scale=1
    comment; do NOT "fix" this to averaging),
  * xGMI is point-to-point (7 links/GPU): grads for the Large model are
    only ~266 MB fp32, so latency and overlap dominate — the default
    bucket size (25 MB) keeps a handful of in-flight collectives without
    fragmenting into launch-bound slivers.
"""

from __future__ import annotations

from typing import List

import torch
import torch.distributed as td


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], comm_dtype):
        self.params = params
        total = sum(p.numel() for p in params)
        dev = params[0].device
        self.flat = torch.zeros(total, dtype=torch.float32, device=dev)
        # wire format: bf16 halves the xGMI bytes (266 MB fp32 grads for
        # the Large model would otherwise rival the 3.4 ms compute step
        # at DP8); fp32 wire via ZAREMBA_AMD_ALLREDUCE_FP32=1
        self.comm_dtype = comm_dtype
        self.comm = (self.flat if comm_dtype == torch.float32 else
                     torch.zeros(total, dtype=comm_dtype, device=dev))
        offset = 0
        for p in params:
            n = p.numel()
            p.grad = self.flat[offset:offset + n].view_as(p)
            offset += n
        self.pending = len(params)
        self.work = None

    def reset(self):
        self.pending = len(self.params)
        self.work = None


class GradBucketer:
    def __init__(self, model: torch.nn.Module, bucket_bytes: int = None,
                 comm_dtype=None):
        if not (td.is_available() and td.is_initialized()):
            raise RuntimeError("GradBucketer requires an initialized process group")
        if bucket_bytes is None:
            import os
            # ZAREMBA_AMD_BUCKET_MB tunes overlap granularity on real
            # multi-GPU nodes (default 25 MB: a handful of in-flight
            # collectives without launch-bound slivers)
            bucket_bytes = int(float(
                os.environ.get("ZAREMBA_AMD_BUCKET_MB", "25")) * (1 << 20))
        if comm_dtype is None:
            import os
            comm_dtype = (torch.float32
                          if os.environ.get("ZAREMBA_AMD_ALLREDUCE_FP32") == "1"
                          or not torch.cuda.is_available()
                          else torch.bfloat16)
        self.world_size = td.get_world_size()
        if td.get_rank() == 0:
            # The wire dtype is a numerics choice (bf16 halves xGMI bytes
            # but rounds the summands); make it visible in every DP log.
            print(f"[zaremba_amd] DP gradient all-reduce wire dtype: "
                  f"{str(comm_dtype).replace('torch.', '')} "
                  f"(ZAREMBA_AMD_ALLREDUCE_FP32=1 forces fp32)")
        params = [p for p in model.parameters() if p.requires_grad]
        params.reverse()  # backward completion order
        self.buckets: List[_Bucket] = []
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        for p in params:
            cur.append(p)
            cur_bytes += p.numel() * 4
            if cur_bytes >= bucket_bytes:
                self.buckets.append(_Bucket(cur, comm_dtype))
                cur, cur_bytes = [], 0
        if cur:
            self.buckets.append(_Bucket(cur, comm_dtype))
        self._by_param = {}
        self._hooks = []
        for b in self.buckets:
            for p in b.params:
                self._by_param[p] = b
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._on_grad_ready))

    def _on_grad_ready(self, param: torch.nn.Parameter):
        b = self._by_param[param]
        b.pending -= 1
        if b.pending == 0:
            if b.comm is not b.flat:
                b.comm.copy_(b.flat)
            b.work = td.all_reduce(b.comm, op=td.ReduceOp.SUM, async_op=True)

    def zero_grad(self):
        """Zero the flat comm buffers (the grads are views into them) and
        re-arm the countdowns. Callers use THIS instead of
        model.zero_grad() under DP: one fill per bucket instead of one
        per parameter (autograd then accumulates into the zeroed
        views)."""
        for b in self.buckets:
            b.flat.zero_()
            b.reset()

    def finalize_backward(self):
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
                if b.comm is not b.flat:
                    b.flat.copy_(b.comm)
            elif b.pending != 0:
                # A parameter produced no grad this step (should not happen in
                # this model); reduce anyway so replicas stay in sync.
                if b.comm is not b.flat:
                    b.comm.copy_(b.flat)
                td.all_reduce(b.comm, op=td.ReduceOp.SUM)
                if b.comm is not b.flat:
                    b.flat.copy_(b.comm)

    def detach_hooks(self):
        for h in self._hooks:
            h.remove()
