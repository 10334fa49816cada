"""Distributed bootstrap + rank helpers (RCCL over xGMI on MI355X).

Capability parity: /root/reference/utils_ret.py:439-523
(init_distributed_mode, rank/world helpers, print gating) — rebuilt for
one-process-per-GPU torch.distributed. Backend: "nccl" (== RCCL on ROCm)
when CUDA devices exist, "gloo" otherwise (CPU tests run world_size>1 on
gloo). SLURM-style env derivation kept (SLURM_PROCID) for parity.
"""
from __future__ import annotations

import builtins
import datetime
import os

import torch
import torch.distributed as dist


def is_dist() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_dist() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_dist() else 1


def get_local_rank() -> int:
    if "LOCAL_RANK" in os.environ:
        return int(os.environ["LOCAL_RANK"])
    return get_rank()


def is_main_process() -> bool:
    return get_rank() == 0


def barrier():
    if is_dist():
        dist.barrier()


def setup_for_distributed(is_master: bool):
    """Gate print() on non-master ranks (reference: utils_ret.py:476-488)."""
    builtin_print = builtins.print

    def print_(*args, **kwargs):
        force = kwargs.pop("force", False)
        if is_master or force:
            builtin_print(*args, **kwargs)

    builtins.print = print_


def init_distributed_mode(backend: str | None = None, timeout_s: int = 1800,
                          gate_print: bool = True) -> tuple[int, int, int]:
    """Initialize from env (torchrun) or SLURM. Returns (rank, world, local)."""
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        rank = int(os.environ["RANK"])
        world = int(os.environ["WORLD_SIZE"])
        local = int(os.environ.get("LOCAL_RANK", rank % max(1, torch.cuda.device_count() or 1)))
    elif "SLURM_PROCID" in os.environ:
        rank = int(os.environ["SLURM_PROCID"])
        world = int(os.environ["SLURM_NTASKS"])
        local = rank % max(1, torch.cuda.device_count() or 1)
    else:
        return 0, 1, 0  # single process

    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if not dist.is_initialized():
        if backend == "nccl":
            torch.cuda.set_device(local)
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
        dist.barrier()
    if gate_print:
        setup_for_distributed(rank == 0)
    return rank, world, local


def all_reduce_mean(x: torch.Tensor) -> torch.Tensor:
    """Mean across ranks (reference: utils_ret.py:877-885)."""
    if not is_dist():
        return x
    y = x.clone()
    dist.all_reduce(y)
    return y / get_world_size()
