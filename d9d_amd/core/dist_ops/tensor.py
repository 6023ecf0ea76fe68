"""Convenience collectives with auto-allocated buffers.

MI355X-native analog of the reference's dist_ops (d9d/core/dist_ops/tensor.py):
thin wrappers over RCCL collectives that allocate receive buffers and handle
tensors whose shapes differ across ranks (two-phase ndim→shape→data exchange).
"""

import torch
import torch.distributed as dist
from torch.distributed import ProcessGroup


def _group_ranks(group: ProcessGroup | None) -> list[int]:
    if group is None:
        return list(range(dist.get_world_size()))
    return dist.get_process_group_ranks(group)


def gather(
    tensor: torch.Tensor,
    dst: int = 0,
    group: ProcessGroup | None = None,
) -> list[torch.Tensor] | None:
    """Gather equal-shape tensors to `dst` (group-relative rank). Returns None elsewhere."""
    group_rank = dist.get_rank(group)
    if group_rank == dst:
        out = [torch.empty_like(tensor) for _ in range(dist.get_world_size(group))]
        dist.gather(tensor, out, dst=dist.get_global_rank(group, dst) if group else dst, group=group)
        return out
    dist.gather(tensor, None, dst=dist.get_global_rank(group, dst) if group else dst, group=group)
    return None


def all_gather(
    tensor: torch.Tensor,
    group: ProcessGroup | None = None,
) -> list[torch.Tensor]:
    """All-gather equal-shape tensors from every rank of the group."""
    out = [torch.empty_like(tensor) for _ in range(dist.get_world_size(group))]
    dist.all_gather(out, tensor.contiguous(), group=group)
    return out


def all_gather_variadic_shape(
    tensor: torch.Tensor,
    group: ProcessGroup | None = None,
) -> list[torch.Tensor]:
    """All-gather tensors whose shapes differ across ranks.

    Two phases: exchange shapes (padded to max ndim), then exchange data into
    per-rank-sized buffers (reference: tensor.py:85).
    """
    world = dist.get_world_size(group)
    device = tensor.device

    ndim = torch.tensor([tensor.dim()], dtype=torch.int64, device=device)
    ndims = [torch.empty_like(ndim) for _ in range(world)]
    dist.all_gather(ndims, ndim, group=group)
    max_ndim = max(int(n.item()) for n in ndims)

    shape = torch.full((max_ndim,), -1, dtype=torch.int64, device=device)
    if tensor.dim() > 0:
        shape[: tensor.dim()] = torch.tensor(tensor.shape, dtype=torch.int64, device=device)
    shapes = [torch.empty_like(shape) for _ in range(world)]
    dist.all_gather(shapes, shape, group=group)

    out: list[torch.Tensor] = []
    flat = tensor.contiguous().view(-1)
    for r in range(world):
        r_shape = tuple(int(s) for s in shapes[r].tolist() if s >= 0)
        numel = 1
        for s in r_shape:
            numel *= s
        buf = torch.empty(numel, dtype=tensor.dtype, device=device)
        out.append((buf, r_shape))

    # Exchange data via per-rank broadcasts (sizes differ, so no single all_gather).
    ranks = _group_ranks(group)
    my_rank = dist.get_rank(group)
    for r, (buf, _) in enumerate(out):
        dist.broadcast(flat if r == my_rank else buf, src=ranks[r], group=group)
    return [
        (flat if r == my_rank else buf).view(r_shape)
        for r, (buf, r_shape) in enumerate(out)
    ]


def gather_variadic_shape(
    tensor: torch.Tensor,
    dst: int = 0,
    group: ProcessGroup | None = None,
) -> list[torch.Tensor] | None:
    """Gather different-shape tensors to `dst` via P2P isend/irecv (reference: tensor.py:113)."""
    world = dist.get_world_size(group)
    my_rank = dist.get_rank(group)
    device = tensor.device
    ranks = _group_ranks(group)

    # Phase 1: shapes to dst via all_gather of padded shape metadata (cheap).
    meta = torch.full((9,), -1, dtype=torch.int64, device=device)
    meta[0] = tensor.dim()
    if tensor.dim() > 0:
        meta[1 : 1 + tensor.dim()] = torch.tensor(tensor.shape, dtype=torch.int64, device=device)
    metas = [torch.empty_like(meta) for _ in range(world)]
    dist.all_gather(metas, meta, group=group)

    if my_rank == dst:
        out: list[torch.Tensor] = []
        works = []
        for r in range(world):
            nd = int(metas[r][0].item())
            r_shape = tuple(int(s) for s in metas[r][1 : 1 + nd].tolist())
            if r == my_rank:
                out.append(tensor.contiguous())
                continue
            buf = torch.empty(r_shape, dtype=tensor.dtype, device=device)
            works.append(dist.irecv(buf, src=ranks[r], group=group))
            out.append(buf)
        for w in works:
            w.wait()
        return out
    dist.send(tensor.contiguous(), dst=ranks[dst], group=group)
    return None
