"""Distributed process-group state: tensor parallelism over RCCL / xGMI.

One process per GPU (torch.distributed; backend "nccl" IS RCCL on ROCm,
"gloo" for CPU tests). TP groups are contiguous rank ranges of size
tensor_parallel_size; replicas (data parallel) are the distinct TP groups.

xGMI note: intra-node links are point-to-point
(7 x ~153 GB/s per GPU); for TP<=8 RCCL's all-reduce over those links is the
right primitive — we keep per-layer all-reduce tensors large (fused qkv and
fused gate-up keep GEMM count and collective count at 2 per layer).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_TP_GROUP: Optional[object] = None
_TP_SIZE = 1
_TP_RANK = 0


def init_distributed(tp_size: int = 1, backend: Optional[str] = None) -> None:
    """Initialise torch.distributed (if needed) and carve out TP groups."""
    global _TP_GROUP, _TP_SIZE, _TP_RANK
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
    if tp_size <= 1 or not dist.is_initialized():
        _TP_GROUP, _TP_SIZE, _TP_RANK = None, 1, 0
        return
    rank = dist.get_rank()
    world = dist.get_world_size()
    assert world % tp_size == 0, "world size must be a multiple of TP size"
    group = None
    for start in range(0, world, tp_size):
        ranks = list(range(start, start + tp_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            group = g
    _TP_GROUP = group
    _TP_SIZE = tp_size
    _TP_RANK = rank % tp_size


def tp_size() -> int:
    return _TP_SIZE


def tp_rank() -> int:
    return _TP_RANK


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    if _TP_SIZE == 1:
        return t
    dist.all_reduce(t, group=_TP_GROUP)
    return t


def tp_all_gather(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    if _TP_SIZE == 1:
        return t
    parts = [torch.empty_like(t) for _ in range(_TP_SIZE)]
    dist.all_gather(parts, t.contiguous(), group=_TP_GROUP)
    return torch.cat(parts, dim=dim)


def destroy() -> None:
    global _TP_GROUP, _TP_SIZE, _TP_RANK
    _TP_GROUP, _TP_SIZE, _TP_RANK = None, 1, 0
