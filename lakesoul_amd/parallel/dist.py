"""torch.distributed helpers.

One process per GPU; backend "nccl" IS RCCL on ROCm (xGMI intra-node).
CPU tests use gloo. Rank auto-detection mirrors the reference's
``python/src/lakesoul/arrow/dataset.py:353-394`` (shard from
torch.distributed if initialized, else env, else single process).
"""

from __future__ import annotations

import os
from typing import Optional, Tuple


def get_rank_world() -> Tuple[int, int]:
    """(rank, world_size) from torch.distributed if initialized, else env."""
    try:
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            return dist.get_rank(), dist.get_world_size()
    except ImportError:  # pragma: no cover
        pass
    return int(os.environ.get("RANK", "0")), int(os.environ.get("WORLD_SIZE", "1"))


def init_from_env(backend: Optional[str] = None):
    """Initialize the default process group from torchrun env vars.

    Returns the torch.distributed module, or None for single-process runs.
    """
    import torch
    import torch.distributed as dist

    if int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        return None
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
        if backend == "nccl":
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    return dist


def barrier(device: Optional[str] = None):
    import torch
    import torch.distributed as dist

    if dist.is_available() and dist.is_initialized():
        dist.barrier()
    if (device is not None and str(device).startswith("cuda")) or (device is None and torch.cuda.is_available()):
        torch.cuda.synchronize()
