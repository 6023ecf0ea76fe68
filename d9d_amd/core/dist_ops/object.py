"""Object collectives (pickle-based; reference: d9d/core/dist_ops/object.py)."""

from typing import Any, TypeVar

import torch.distributed as dist
from torch.distributed import ProcessGroup

T = TypeVar("T")


def gather_object(
    obj: T,
    dst: int = 0,
    group: ProcessGroup | None = None,
) -> list[T] | None:
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    global_dst = dist.get_global_rank(group, dst) if group else dst
    out: list[Any] | None = [None] * world if rank == dst else None
    dist.gather_object(obj, out, dst=global_dst, group=group)
    return out


def all_gather_object(
    obj: T,
    group: ProcessGroup | None = None,
) -> list[T]:
    world = dist.get_world_size(group)
    out: list[Any] = [None] * world
    dist.all_gather_object(out, obj, group=group)
    return out
