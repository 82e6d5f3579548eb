"""torch.distributed bootstrap: one process per GPU over RCCL.

On ROCm the "nccl" backend IS RCCL, riding the 7x153 GB/s point-to-point
xGMI mesh of one 8xMI355X node; CPU tests use gloo.
Single-process runs never initialize a process group — every helper
degrades to world_size=1.
"""

from __future__ import annotations

import os
from datetime import timedelta
from typing import Optional

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def get_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", get_rank()))


def init_distributed(backend: Optional[str] = None, timeout_s: int = 600) -> int:
    """Initialize from torchrun env vars; returns world size.

    backend default: "nccl" (=RCCL) when GPUs are visible, else "gloo".
    """
    if is_distributed():
        return get_world_size()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 1
    if backend is None:
        backend = os.environ.get("DTS_DIST_BACKEND") or (
            "nccl" if torch.cuda.is_available() else "gloo"
        )
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    dist.init_process_group(backend=backend, timeout=timedelta(seconds=timeout_s))
    if torch.cuda.is_available() and backend == "nccl":
        torch.cuda.set_device(get_local_rank())
    return get_world_size()


def barrier() -> None:
    if is_distributed():
        dist.barrier()


def destroy() -> None:
    if is_distributed():
        dist.destroy_process_group()
