"""DP-replicate (DDP) parallelization (reference: d9d/module/parallelism/api/replicate_parallel.py:9).

Parameters become DTensor-Replicate over the given mesh; gradient averaging is
NOT done here — the bucketed `GradientSynchronizer`
(d9d_amd/internals/grad_sync) all-reduces over every Replicate mesh dim on a
dedicated HIP stream, overlapped with backward.
"""

import torch
import torch.distributed as dist
from torch import nn
from torch.distributed.device_mesh import DeviceMesh
from torch.distributed.tensor import DTensor

from .style import distribute_module_params


def parallelize_replicate(
    module: nn.Module, mesh: DeviceMesh, broadcast_params: bool = True
) -> nn.Module:
    """Replicate every parameter of `module` over all dims of `mesh`.

    `broadcast_params` copies group-rank-0's values over each mesh dim so
    replicas agree even when the caller seeded per-rank (DDP semantics);
    pass False when weights are known-identical (e.g. loaded checkpoint).

    A "pp" dim in the mesh is dropped automatically: pipeline stages hold
    DIFFERENT modules, so parameters are never replicated across pp (the
    dense/batch domains carry pp for other consumers).
    """
    if mesh.mesh_dim_names and "pp" in mesh.mesh_dim_names and mesh.ndim > 1:
        keep = tuple(n for n in mesh.mesh_dim_names if n != "pp")
        mesh = mesh[keep]
    distribute_module_params(module, mesh, placement_fn=None, recurse=True)
    if broadcast_params:
        # Only tensors replicated over THIS mesh: params parallelized earlier
        # by another strategy (e.g. EP-sharded experts on the expert mesh)
        # hold different data per rank by design and must not be overwritten.
        def _broadcastable(t):
            if not isinstance(t, DTensor):
                return True
            return t.device_mesh is mesh

        def _do_broadcast(mod):
            with torch.no_grad():
                for dim in range(mesh.ndim):
                    group = mesh.get_group(dim)
                    if dist.get_world_size(group) == 1:
                        continue
                    src_rank = dist.get_global_rank(group, 0)
                    for t in list(mod.parameters()) + list(mod.buffers()):
                        if not _broadcastable(t):
                            continue
                        local = t.to_local() if isinstance(t, DTensor) else t
                        dist.broadcast(local, src=src_rank, group=group)

        def _is_meta(t):
            # a DTensor over a cpu/cuda mesh whose LOCAL tensor is meta
            # reports wrapper-level is_meta False — check the local
            lt = getattr(t, "_local_tensor", None)
            if lt is not None:
                return lt.is_meta
            return t.is_meta

        tensors = list(module.parameters()) + list(module.buffers())
        if any(_is_meta(t) for t in tensors):
            # meta-device build flow: real values exist only after
            # to_empty + reset_parameters — defer the broadcast there
            cbs = getattr(module, "_d9d_post_materialize", None)
            if cbs is None:
                cbs = []
                module._d9d_post_materialize = cbs
            cbs.append(_do_broadcast)
        else:
            _do_broadcast(module)
    return module
