"""torch.distributed bootstrap helpers.

One process per GPU; RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* read from the
environment (torchrun contract). Backend "nccl" IS RCCL on ROCm; tests use
"gloo" on CPU with the identical collective call pattern.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch
import torch.distributed as td


def is_distributed() -> bool:
    return td.is_available() and td.is_initialized()


def get_world() -> Tuple[int, int]:
    """(rank, world_size), valid with or without init."""
    if is_distributed():
        return td.get_rank(), td.get_world_size()
    return 0, 1


def init_from_env(backend: Optional[str] = None) -> Tuple[int, int, torch.device]:
    """Initialise the process group from torchrun env vars.

    Returns (rank, world_size, device). Single-process callers (WORLD_SIZE
    unset or 1) get (0, 1, best-device) without initialising a group.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        dev = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
        return 0, 1, dev

    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not td.is_initialized():
        td.init_process_group(backend=backend)
    rank = td.get_rank()
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if torch.cuda.is_available():
        device = torch.device(f"cuda:{local_rank}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    return rank, td.get_world_size(), device
