"""Distributed helpers: one process per GPU over RCCL/xGMI.

Replaces the reference's gRPC manager/worker distribution
(utils/distribute/, distributed_gradient_boosted_trees.cc): the only
collective the training algorithm needs is the per-level histogram
all-reduce (+ scalar stat reductions), done with torch.distributed
(backend "nccl" IS RCCL on ROCm; "gloo" for CPU tests).
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_from_env(backend: str = None) -> int:
    """Initializes torch.distributed from torchrun env vars; returns rank.

    No-op (returns 0) when not launched by torchrun (RANK absent). A
    world-size-1 torchrun launch DOES initialize a process group, so the
    full RCCL init + collective path runs in single-GPU rehearsals."""
    if "RANK" not in os.environ or "MASTER_ADDR" not in os.environ:
        return 0
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if not dist.is_initialized():
        if backend is None:
            backend = os.environ.get("YDFA_DIST_BACKEND")
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            n_dev = max(torch.cuda.device_count(), 1)
            if world > n_dev:
                # RCCL refuses two ranks on one device ("Duplicate GPU
                # detected"); fall back to gloo so over-subscribed
                # rehearsal launches still run end-to-end
                import sys

                print(f"# ydf_amd: {world} ranks > {n_dev} visible "
                      f"GPU(s); using gloo backend for this rehearsal",
                      file=sys.stderr)
                backend = "gloo"
            else:
                torch.cuda.set_device(
                    int(os.environ.get("LOCAL_RANK", "0")) % n_dev)
        dist.init_process_group(backend=backend,
                                timeout=datetime.timedelta(seconds=300))
    return dist.get_rank()


def is_main() -> bool:
    return (not dist.is_available()) or (not dist.is_initialized()) \
        or dist.get_rank() == 0


def world_size() -> int:
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size()
    return 1


def barrier() -> None:
    if dist.is_available() and dist.is_initialized():
        dist.barrier()


def shard_rows(n: int, rank: int = None, world: int = None):
    """Contiguous row shard [lo, hi) for this rank."""
    if world is None:
        world = world_size()
    if rank is None:
        rank = dist.get_rank() if world > 1 else 0
    per = (n + world - 1) // world
    lo = min(rank * per, n)
    hi = min(lo + per, n)
    return lo, hi
