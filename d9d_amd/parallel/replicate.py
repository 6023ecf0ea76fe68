"""DP-replicate (DDP) parallelization (reference: d9d/module/parallelism/api/replicate_parallel.py:9).

Parameters become DTensor-Replicate over the given mesh; gradient averaging is
NOT done here — the bucketed `GradientSynchronizer`
(d9d_amd/internals/grad_sync) all-reduces over every Replicate mesh dim on a
dedicated HIP stream, overlapped with backward.
"""

from torch import nn
from torch.distributed.device_mesh import DeviceMesh

from .style import distribute_module_params


def parallelize_replicate(module: nn.Module, mesh: DeviceMesh) -> nn.Module:
    """Replicate every parameter of `module` over all dims of `mesh`."""
    return distribute_module_params(module, mesh, placement_fn=None, recurse=True)
