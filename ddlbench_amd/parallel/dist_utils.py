"""torch.distributed bootstrap — one process per GPU over RCCL.

Replaces the reference's horovodrun/hvd.init() bootstrap
(/root/reference/benchmark/mnist/mnist_horovod.py:163-171) and pipedream's
dist.init_process_group (runtime/communication.py:40-46) with one
torchrun-compatible helper: ranks read RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*
from the environment; backend "nccl" IS RCCL on ROCm."""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class DistEnv:
    rank: int
    local_rank: int
    world_size: int

    @property
    def is_master(self) -> bool:
        return self.rank == 0


def distributed_env() -> DistEnv:
    return DistEnv(
        rank=int(os.environ.get("RANK", "0")),
        local_rank=int(os.environ.get("LOCAL_RANK", "0")),
        world_size=int(os.environ.get("WORLD_SIZE", "1")),
    )


def init_distributed(backend: str = "auto",
                     timeout_s: int = 300) -> DistEnv:
    """Initialize the default process group from torchrun env vars.

    backend "auto": RCCL ("nccl") when HIP devices exist, else gloo."""
    env = distributed_env()
    if env.world_size <= 1:
        return env
    if backend == "auto":
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if not dist.is_initialized():
        if backend == "nccl":
            torch.cuda.set_device(env.local_rank)
        dist.init_process_group(
            backend=backend, rank=env.rank, world_size=env.world_size,
            timeout=datetime.timedelta(seconds=timeout_s))
    return env


def allreduce_mean_scalar(value: float, device=None) -> float:
    """Average a python scalar across ranks (metric averaging — the
    reference's hvd.allreduce, mnist_horovod.py:129-132)."""
    if not (dist.is_available() and dist.is_initialized()):
        return value
    if device is None:
        device = (torch.device("cuda", torch.cuda.current_device())
                  if dist.get_backend() == "nccl" else torch.device("cpu"))
    t = torch.tensor([value], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return (t / dist.get_world_size()).item()


def barrier() -> None:
    if dist.is_available() and dist.is_initialized():
        dist.barrier()
