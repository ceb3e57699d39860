"""Distributed setup: one process per GPU over RCCL/xGMI.

The reference has no distributed backend; this framework
scales training data-parallel with ``torch.distributed`` — backend
``"nccl"`` IS RCCL on ROCm — and falls back to ``gloo`` on CPU (which is
how the multi-process paths are tested without GPUs).

Communication volumes are tiny by design: the OLS fit
all-reduces five fp64 scalars; the MLP fit all-reduces one flat fp32
gradient bucket per step (~64 MiB for the 4096-d config) — a single RCCL
launch, latency- not bandwidth-bound on 7x153 GB/s xGMI links.
"""
from __future__ import annotations

import os
from contextlib import contextmanager
from datetime import timedelta

import torch
import torch.distributed as dist

from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)


def init_distributed(backend: str | None = None, timeout_s: float = 300.0):
    """Initialise from torchrun env vars; returns (rank, world, local_rank).

    No-op (0, 1, 0) when WORLD_SIZE is absent or 1.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1, 0
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if backend is None:
        # RCCL refuses two ranks on one device ("Duplicate GPU detected",
        # ncclInvalidUsage — profiles/r02_world2_rccl.md), so NCCL/RCCL is
        # only auto-selected when every rank can own a distinct GPU; an
        # oversubscribed world (the 1-GPU world=2 rehearsal) falls back to
        # gloo transport with the compute still on the GPU.
        n_gpu = torch.cuda.device_count() if torch.cuda.is_available() else 0
        backend = "nccl" if 0 < world <= n_gpu else "gloo"
    if backend == "nccl":
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend, rank=rank, world_size=world,
            timeout=timedelta(seconds=timeout_s),
        )
        log.info(f"initialised {backend} rank {rank}/{world} "
                 f"(local_rank={local_rank})")
    return rank, world, local_rank


@contextmanager
def distributed_context(backend: str | None = None):
    rank, world, local_rank = init_distributed(backend)
    try:
        yield rank, world, local_rank
    finally:
        if dist.is_initialized():
            dist.barrier()
            dist.destroy_process_group()
