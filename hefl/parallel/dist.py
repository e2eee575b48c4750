"""torch.distributed process-group plumbing: the federation fabric.

1 rank = 1 FL client (= 1 MI355X GPU on the nccl/RCCL backend). Replaces the
reference's pickle-files-on-disk transport (FLPyfhelin.py:230-240, :303-328)
with RCCL collectives over xGMI (GPU) or gloo (CPU tests).
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None, timeout_s: int = 300) -> int:
    """Initialize from torchrun env vars; no-op for single-process runs.

    Returns the local rank. Sets the CUDA device for nccl (=RCCL on ROCm).
    """
    if dist.is_initialized():
        return int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    local_rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))
    if backend == "nccl":
        torch.cuda.set_device(local_rank)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(seconds=timeout_s))
    return local_rank


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def barrier():
    if dist.is_initialized():
        if torch.cuda.is_available() and dist.get_backend() == "nccl":
            dist.barrier(device_ids=[torch.cuda.current_device()])
        else:
            dist.barrier()
