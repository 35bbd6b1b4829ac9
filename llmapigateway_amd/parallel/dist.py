"""torch.distributed bootstrap for MI355X multi-GPU (RCCL over xGMI).

One process per GPU; backend "nccl" IS RCCL on ROCm. All 8 GPUs are
intra-node (fully-connected 7-link xGMI mesh), so there is no multi-node
path (SURVEY.md §5 "Distributed communication backend").
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch
import torch.distributed as dist


def init_from_env(backend: Optional[str] = None) -> Tuple[int, int]:
    """Init the default process group from torchrun env; returns (rank, world)."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world <= 1:
        return 0, 1
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    return rank, world


def get_tp_info(tp_group=None) -> Tuple[int, int]:
    if not dist.is_initialized():
        return 0, 1
    return dist.get_rank(tp_group), dist.get_world_size(tp_group)
