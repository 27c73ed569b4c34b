"""Distributed bootstrap: torchrun env -> RCCL (nccl backend on ROCm) / gloo.

Replaces the reference's TF_CONFIG cluster spec + gRPC rendezvous
(03:68-74, 04:98-104) with torch.distributed process groups. One process per
GPU; RCCL rides xGMI intra-node.
"""

from __future__ import annotations

import os
from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class DistContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    device: torch.device = torch.device("cpu")

    @property
    def is_main(self) -> bool:
        return self.rank == 0


def init_distributed(backend: str = "auto") -> DistContext:
    """Initialize from torchrun env vars; no-op single-process otherwise."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        if backend == "auto":
            backend = "nccl" if use_cuda else "gloo"
        dist.init_process_group(backend, rank=rank, world_size=world)
    return DistContext(rank=rank, world_size=world, local_rank=local_rank, device=device)


def cleanup() -> None:
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()
