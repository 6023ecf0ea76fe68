"""Distributed gradient-norm clipping (reference: d9d/internals/grad_norm/).

Local foreach squared norms, async all-reduce of pow-norms over each shard
mesh (Shard/Partial dims), then over pp, then one foreach scale.
"""

import torch
import torch.distributed as dist
from torch.distributed.tensor import DTensor, Replicate


def _shard_groups(param: torch.Tensor):
    """Process groups whose ranks hold DISJOINT pieces of this param's grad
    (Shard placements) — the pow-norm must be summed over them."""
    if not isinstance(param, DTensor):
        return []
    mesh = param.device_mesh
    return [
        mesh.get_group(d)
        for d, placement in enumerate(param.placements)
        if not isinstance(placement, Replicate) and mesh.shape[d] > 1
    ]


def clip_grad_norm_distributed_(
    parameters,
    max_norm: float,
    pp_group=None,
    foreach: bool = True,
) -> torch.Tensor:
    """Clip in place; returns the total grad norm (fp32, on the grads' device)."""
    params = [p for p in parameters if p.grad is not None]
    if not params:
        return torch.tensor(0.0)

    grads_local = [
        (p.grad.to_local() if isinstance(p.grad, DTensor) else p.grad)
        for p in params
    ]
    device = grads_local[0].device

    # Group params by their shard-group signature so each signature's local
    # pow-norm is reduced over the right groups exactly once.
    sig_to_groups: dict[tuple, list] = {}
    sig_to_sq: dict[tuple, torch.Tensor] = {}
    for p, g in zip(params, grads_local):
        groups = _shard_groups(p)
        sig = tuple(id(gr) for gr in groups)
        sq = g.float().pow(2).sum()
        if sig in sig_to_sq:
            sig_to_sq[sig] = sig_to_sq[sig] + sq
        else:
            sig_to_sq[sig] = sq
            sig_to_groups[sig] = groups

    total_sq = torch.zeros((), dtype=torch.float32, device=device)
    for sig, sq in sig_to_sq.items():
        for group in sig_to_groups[sig]:
            dist.all_reduce(sq, group=group)
        total_sq += sq

    if pp_group is not None:
        dist.all_reduce(total_sq, group=pp_group)

    total_norm = total_sq.sqrt()
    clip_coef = (max_norm / (total_norm + 1e-6)).clamp(max=1.0)
    if foreach:
        torch._foreach_mul_(grads_local, clip_coef)
    else:
        for g in grads_local:
            g.mul_(clip_coef)
    return total_norm
