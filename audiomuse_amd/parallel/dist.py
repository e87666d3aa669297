"""torch.distributed helpers: one process per GPU over RCCL.

The deployment shape: DP=8 on one node, backend
"nccl" (RCCL on ROCm) over xGMI; CPU tests use gloo. Rendezvous always
binds 127.0.0.1 (container hostnames may not resolve).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


def init_from_env(backend: Optional[str] = None) -> int:
    """Initialize from torchrun env; returns local_rank (no-op world=1)."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        backend = backend or ("nccl" if torch.cuda.is_available() else "gloo")
        dist.init_process_group(backend=backend,
                                rank=int(os.environ.get("RANK", "0")),
                                world_size=world)
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return local_rank


def world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def is_main() -> bool:
    return rank() == 0


def barrier() -> None:
    if dist.is_initialized():
        dist.barrier()


def cleanup() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
