"""FSDP / HSDP parallelization (reference: d9d/module/parallelism/api/fully_sharded.py:8-41,
hybrid_sharded.py:10-43).

Wraps torch's composable `fully_shard` (FSDP2 — all-gather forward /
reduce-scatter backward over RCCL) with the reference's overrides: SUM
reduction with divide factor 1 so the trainer scales gradients itself
(loss-weighted accumulation), matching the GradientSynchronizer convention.
"""

from torch import nn
from torch.distributed.device_mesh import DeviceMesh
from torch.distributed.fsdp import fully_shard
from torch.distributed.tensor import DTensor


def _foreign_dtensor_params(module: nn.Module, mesh: DeviceMesh) -> set[nn.Parameter]:
    """Params already DTensor-sharded on ANOTHER mesh (e.g. EP experts on the
    expert domain, TP shards on the regular domain): fully_shard cannot
    compose overlapping meshes, so these are ignored — their cross-replica
    grad sync rides the GradientSynchronizer's Replicate-dim reduction."""
    out: set[nn.Parameter] = set()
    for p in module.parameters():
        if isinstance(p, DTensor) and p.device_mesh is not mesh:
            out.add(p)
    return out


def _apply_sum_reduction(module: nn.Module) -> None:
    # FSDP2 defaults to averaging over the shard group; d9d semantics are SUM
    # (the trainer divides by the accumulated loss weight itself). Forcing
    # plain-SUM comms makes the divide factor a host-side scale instead of an
    # NCCL premul-sum reduce op, which gloo (the CPU test backend) lacks.
    set_force = getattr(module, "set_force_sum_reduction_for_comms", None)
    if callable(set_force):
        set_force(True)
    set_factor = getattr(module, "set_gradient_divide_factor", None)
    if callable(set_factor):
        set_factor(1.0)


def parallelize_fsdp(
    module: nn.Module,
    mesh: DeviceMesh,
    reshard_after_forward: bool = True,
    shard_units: list[nn.Module] | None = None,
) -> nn.Module:
    """Shard parameters over a 1-D mesh.

    `shard_units`: inner modules to shard as their own FSDP groups (e.g.
    decoder layers) so all-gathers pipeline with compute; the root module is
    always wrapped last.
    """
    assert mesh.ndim == 1, "parallelize_fsdp expects a 1-D mesh"
    ignored = _foreign_dtensor_params(module, mesh)
    for unit in shard_units or []:
        fully_shard(unit, mesh=mesh, reshard_after_forward=reshard_after_forward,
                    ignored_params=ignored)
        _apply_sum_reduction(unit)
    fully_shard(module, mesh=mesh, reshard_after_forward=reshard_after_forward,
                ignored_params=ignored)
    _apply_sum_reduction(module)
    return module


def parallelize_hsdp(
    module: nn.Module,
    mesh: DeviceMesh,
    reshard_after_forward: bool = True,
    shard_units: list[nn.Module] | None = None,
) -> nn.Module:
    """Hybrid sharding over a 2-D (replicate, shard) mesh."""
    assert mesh.ndim == 2, "parallelize_hsdp expects a 2-D (replicate, shard) mesh"
    ignored = _foreign_dtensor_params(module, mesh)
    for unit in shard_units or []:
        fully_shard(unit, mesh=mesh, reshard_after_forward=reshard_after_forward,
                    ignored_params=ignored)
        _apply_sum_reduction(unit)
    fully_shard(module, mesh=mesh, reshard_after_forward=reshard_after_forward,
                ignored_params=ignored)
    _apply_sum_reduction(module)
    return module
