"""Parameter-placement styles.

`ToLocalParallel` (reference: d9d/module/parallelism/style/to_local.py):
parameters are stored as DTensors (so checkpointing, grad sync and the
optimizer see placements) while forward code transparently reads the local
shard — each parameter name becomes a class property returning
`param.to_local(grad_placements=...)`, so gradients flow back into the
DTensor parameter with the declared placements.
"""

from typing import Sequence

import torch
from torch import nn
from torch.distributed.device_mesh import DeviceMesh
from torch.distributed.tensor import DTensor, Placement, Replicate, Shard

_CLASS_CACHE: dict[tuple, type] = {}


def _make_local_property(name: str):
    def getter(self):
        p = self._parameters.get(name)
        if p is None:
            raise AttributeError(name)
        if isinstance(p, DTensor):
            if torch.is_grad_enabled() and p.requires_grad:
                return p.to_local(grad_placements=p.placements)
            # no-grad context (init, inference): raw local shard, in-place ok
            return p._local_tensor
        return p

    return property(getter)


def _to_local_class(cls: type, param_names: tuple[str, ...]) -> type:
    key = (cls, param_names)
    if key not in _CLASS_CACHE:
        ns = {name: _make_local_property(name) for name in param_names}
        ns["_d9d_to_local_params"] = param_names
        _CLASS_CACHE[key] = type(f"ToLocal{cls.__name__}", (cls,), ns)
    return _CLASS_CACHE[key]


def distribute_module_params(
    module: nn.Module,
    mesh: DeviceMesh,
    placement_fn=None,
    recurse: bool = True,
) -> nn.Module:
    """DTensor-ize every direct parameter of `module` (and children if
    `recurse`), swapping each owning module's class so forward reads locals.

    `placement_fn(fqn, param) -> Sequence[Placement] | None` decides the
    placements (None = Replicate over all mesh dims).
    """
    mods = module.modules() if recurse else [module]
    for sub in mods:
        names = [n for n, p in sub.named_parameters(recurse=False)]
        if not names:
            continue
        for name in names:
            p = sub._parameters[name]
            if p is None or isinstance(p, DTensor):
                continue
            placements: Sequence[Placement] | None = None
            if placement_fn is not None:
                placements = placement_fn(name, p)
            if placements is None:
                placements = [Replicate()] * mesh.ndim
            dt = DTensor.from_local(p.data, mesh, tuple(placements), run_check=False)
            new_p = nn.Parameter(dt, requires_grad=p.requires_grad)
            sub._parameters[name] = new_p
        if not getattr(sub, "_d9d_to_local_params", None):
            sub.__class__ = _to_local_class(type(sub), tuple(names))
    return module


def shard_param_dim0(
    module: nn.Module,
    param_name: str,
    mesh: DeviceMesh,
    shard_mesh_dim: int,
) -> None:
    """Shard one parameter on dim 0 over `shard_mesh_dim` (others Replicate);
    the LOCAL tensor is assumed to already hold this rank's shard."""
    p = module._parameters[param_name]
    assert not isinstance(p, DTensor)
    placements: list[Placement] = [Replicate()] * mesh.ndim
    placements[shard_mesh_dim] = Shard(0)
    dt = DTensor.from_local(p.data, mesh, tuple(placements), run_check=False)
    module._parameters[param_name] = nn.Parameter(dt, requires_grad=p.requires_grad)
    if not getattr(module, "_d9d_to_local_params", None):
        names = tuple(n for n, _ in module.named_parameters(recurse=False))
        module.__class__ = _to_local_class(type(module), names)
