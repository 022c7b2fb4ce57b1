"""Communicator bootstrap: one process per GPU over RCCL (or gloo on CPU).

Reads RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT from the
environment (torchrun contract). CPU multi-process tests use the gloo
backend with the same code path (SURVEY.md §5 lesson: the full task path
must run GPU-less in pytest).
"""
from __future__ import annotations

import datetime
import os
from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class CommContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    device: torch.device = torch.device("cpu")
    backend: str = "none"

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1 and dist.is_initialized()

    def barrier(self) -> None:
        if self.is_distributed:
            if self.device.type == "cuda":
                dist.barrier(device_ids=[self.local_rank])
            else:
                dist.barrier()

    def all_reduce(self, tensor: torch.Tensor, op=None, async_op: bool = False):
        if not self.is_distributed:
            return None
        return dist.all_reduce(tensor, op=op or dist.ReduceOp.SUM, async_op=async_op)

    def broadcast(self, tensor: torch.Tensor, src: int = 0, async_op: bool = False):
        if not self.is_distributed:
            return None
        return dist.broadcast(tensor, src=src, async_op=async_op)

    def reduce(self, tensor: torch.Tensor, dst: int = 0, op=None, async_op: bool = False):
        if not self.is_distributed:
            return None
        return dist.reduce(tensor, dst=dst, op=op or dist.ReduceOp.SUM, async_op=async_op)


def init_from_env(force_backend: str | None = None, timeout_s: float = 300.0) -> CommContext:
    """Initialize torch.distributed from torchrun env vars (no-op if absent)."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    cuda = torch.cuda.is_available()
    if cuda:
        torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
        device = torch.device("cuda", local_rank % max(1, torch.cuda.device_count()))
    else:
        device = torch.device("cpu")
    backend = force_backend or ("nccl" if cuda else "gloo")  # nccl IS RCCL on ROCm
    if world > 1 and not dist.is_initialized():
        # host driver supports only dmabuf IPC — without this RCCL's
        # cross-process CUDA-tensor sharing fails with
        # hipIpcGetMemHandle: invalid argument
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29512")
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    return CommContext(rank=rank, world_size=world, local_rank=local_rank,
                       device=device, backend=backend if world > 1 else "none")
