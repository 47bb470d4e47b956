"""Distributed process-group setup: one process per GPU over RCCL/xGMI.

Replaces the reference's Spark driver/executor scheduling (SURVEY.md §1.2
Scheduler row). Two launch modes:
  - torchrun-style: RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* read from the env
    (what the bench driver uses), or
  - self-spawned: torch.multiprocessing.spawn for tests.
Backend "nccl" IS RCCL on ROCm; "gloo" is the CPU test backend.
"""
from __future__ import annotations

import datetime
import os
from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class DistContext:
    rank: int = 0
    local_rank: int = 0
    world_size: int = 1
    backend: str = ""
    device: torch.device = torch.device("cpu")

    @property
    def is_main(self) -> bool:
        return self.rank == 0


def pick_backend(requested: str = "auto") -> str:
    if requested != "auto":
        return requested
    return "nccl" if torch.cuda.is_available() else "gloo"


def init_from_env(backend: str = "auto", timeout_s: int = 300) -> DistContext:
    """Initialize torch.distributed from torchrun-style env vars.

    With no WORLD_SIZE in the env (single-process run) returns a world_size=1
    context without initializing a process group.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if torch.cuda.is_available():
        device = torch.device("cuda", local_rank % torch.cuda.device_count())
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    if world == 1:
        return DistContext(0, 0, 1, "", device)
    be = pick_backend(backend)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(
        backend=be, rank=rank, world_size=world,
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    return DistContext(rank, local_rank, world, be, device)


def cleanup():
    if dist.is_initialized():
        dist.destroy_process_group()


def run_spawned(fn, world_size: int, backend: str = "gloo", args=(),
                master_port: int = 29511):
    """Spawn `world_size` processes running fn(ctx, *args). CPU/gloo path for
    the distributed-semantics test tier."""
    import torch.multiprocessing as mp

    mp.spawn(
        _spawn_entry,
        args=(world_size, backend, master_port, fn, args),
        nprocs=world_size,
        join=True,
    )


def _spawn_entry(rank: int, world_size: int, backend: str, master_port: int,
                 fn, args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(master_port)
    dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
    device = torch.device("cpu")
    if backend == "nccl":
        device = torch.device("cuda", rank)
        torch.cuda.set_device(device)
    ctx = DistContext(rank, rank, world_size, backend, device)
    try:
        fn(ctx, *args)
    finally:
        cleanup()
