"""Process-group plumbing: one process per GPU over RCCL (xGMI intra-node).

The reference is single-process / single-device (no distributed code at
all). Here every rank binds one GPU; the process group is
initialized from the torchrun environment (RANK/WORLD_SIZE/LOCAL_RANK/
MASTER_*). Backend: "nccl" (RCCL on ROCm) when CUDA devices are visible,
"gloo" otherwise (the CPU CI path).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as td


def env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def maybe_init():
    """Initialize the default process group when launched by torchrun.

    Test overrides: ZAREMBA_AMD_PG_BACKEND forces the backend (e.g. gloo
    to exercise the DP code path on one GPU) and ZAREMBA_AMD_ONE_GPU=1
    pins every rank to device 0 (multi-rank smoke tests on a single-GPU
    box)."""
    if env_world_size() <= 1 or td.is_initialized():
        return
    backend = os.environ.get(
        "ZAREMBA_AMD_PG_BACKEND",
        "nccl" if torch.cuda.is_available() else "gloo")
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank())
    td.init_process_group(backend=backend,
                          timeout=datetime.timedelta(seconds=300))


def initialized() -> bool:
    return td.is_available() and td.is_initialized()


def rank() -> int:
    return td.get_rank() if initialized() else 0


def world_size() -> int:
    return td.get_world_size() if initialized() else 1


def local_rank() -> int:
    if os.environ.get("ZAREMBA_AMD_ONE_GPU", "0") == "1":
        return 0
    return int(os.environ.get("LOCAL_RANK", "0"))


def is_rank0() -> bool:
    return rank() == 0


def broadcast_parameters(model: torch.nn.Module, src: int = 0):
    """Make every rank start from rank 0's initialization (the reference
    seeds nothing; DP requires identical replicas)."""
    if not initialized():
        return
    with torch.no_grad():
        for p in model.parameters():
            td.broadcast(p.data, src=src)


def barrier():
    if initialized():
        td.barrier()


def finalize():
    if initialized():
        td.barrier()
        td.destroy_process_group()
