"""Process-group bootstrap for the mi355x harness.

One process per GPU over torch.distributed — the "nccl" backend IS RCCL on
ROCm, riding the 7 point-to-point xGMI links per MI355X. CPU-only runs and
tests use the "gloo" backend with the identical call surface, so the
reducer/sampler logic is exercised without a GPU.

The env contract matches the reference's launcher expectation
(/root/reference/cifar_example_ddp.py:43-58): RANK / LOCAL_RANK /
WORLD_SIZE read from the environment, MASTER_ADDR/MASTER_PORT defaulted to
127.0.0.1:29500 (single node).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


def is_distributed_env() -> bool:
    return "RANK" in os.environ and "WORLD_SIZE" in os.environ


def init_process_group(backend: str | None = None, timeout_s: int = 300):
    """Initialise the default process group from the env contract.

    Returns (rank, world_size, local_rank). Safe to call in a single,
    launcher-less process (world_size 1): still creates a 1-rank group so
    all collective call sites work unchanged.
    """
    rank, world, local = env_rank(), env_world_size(), env_local_rank()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if backend is None:
        # MI355X_BACKEND=gloo lets multi-rank rehearsals share one GPU
        # (RCCL refuses two ranks on the same device)
        backend = os.environ.get(
            "MI355X_BACKEND",
            "nccl" if torch.cuda.is_available() else "gloo")
    if backend == "nccl":
        torch.cuda.set_device(local % max(torch.cuda.device_count(), 1))
    if not dist.is_initialized():
        # MI355X_RDZV_FILE: filesystem-store rendezvous instead of the
        # env:// TCPStore. Used by the multi-process CPU tests — freeing a
        # probed port and rebinding it races with the OS handing the same
        # ephemeral port to any other socket (observed ~1-in-10
        # ConnectionError flakes under repeated suite runs); a file store
        # has no port to lose.
        store_path = os.environ.get("MI355X_RDZV_FILE")
        if store_path:
            store = dist.FileStore(store_path, world)
            dist.init_process_group(backend=backend, store=store, rank=rank,
                                    world_size=world,
                                    timeout=datetime.timedelta(seconds=timeout_s))
        else:
            dist.init_process_group(backend=backend, rank=rank,
                                    world_size=world,
                                    timeout=datetime.timedelta(seconds=timeout_s))
    return rank, world, local


def barrier():
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.barrier()


def world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def destroy():
    if dist.is_initialized():
        dist.destroy_process_group()
