"""Context parallelism (CP) — long-sequence sharding.

The reference plumbs cp mesh dims everywhere but implements no CP
(models raise for cp > 1). Here: the batch's sequence is split
over the `cp_shard` mesh dim; each attention block all-gathers K/V along the
sequence over RCCL (differentiable: backward reduce-scatters dK/dV) and runs
the CDNA4 flash kernel on its local Q block with the global `q_offset`, so
causal masking is exact. LSE stays local (each rank's rows see the full
prefix), no online merge needed.
"""

import torch
from torch import nn
from torch.distributed.device_mesh import DeviceMesh

from ..module.block.attention import GroupedQueryAttention


def parallelize_context_parallel(
    module: nn.Module,
    mesh: DeviceMesh,
    cp_dim_name: str = "cp_shard",
) -> nn.Module:
    cp_dim = mesh.mesh_dim_names.index(cp_dim_name)
    cp_size = mesh.shape[cp_dim]
    if cp_size == 1:
        return module
    group = mesh.get_group(cp_dim)
    cp_rank = mesh.get_coordinate()[cp_dim]
    for sub in module.modules():
        if isinstance(sub, GroupedQueryAttention):
            sub._cp_group = group
            sub._cp_rank = cp_rank
            sub._cp_size = cp_size
    return module


def shard_sequence(batch: torch.Tensor, cp_rank: int, cp_size: int, dim: int = 1) -> torch.Tensor:
    """Slice this rank's sequence chunk (use with position_ids offset)."""
    chunks = batch.chunk(cp_size, dim=dim)
    return chunks[cp_rank].contiguous()
