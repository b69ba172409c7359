"""Parallelism layer: RCCL-over-xGMI data parallelism.

One process per GPU over ``torch.distributed`` (backend "nccl" IS RCCL on ROCm):
  * GradReducer — bucketed all-reduce overlapped with backward
  * SyncBatchNorm2d / convert_syncbn — collective BN statistics
  * init_distributed — env:// process-group bring-up
"""
import os

import torch
import torch.distributed as dist

from .ddp import GradReducer, reduce_tensor
from .syncbn import SyncBatchNorm2d, convert_syncbn


def init_distributed(backend: str | None = None):
    """Initialise the process group from torchrun env vars; returns
    (rank, local_rank, world_size). Single-process mode if WORLD_SIZE unset/1."""
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        return 0, 0, 1
    rank = int(os.environ["RANK"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, init_method="env://",
                                world_size=world_size, rank=rank)
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return rank, local_rank, world_size


__all__ = ["GradReducer", "reduce_tensor", "SyncBatchNorm2d", "convert_syncbn",
           "init_distributed"]
