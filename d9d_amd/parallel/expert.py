"""Expert parallelism (reference: d9d/module/parallelism/api/expert_parallel.py:9-45).

GroupedLinear weights become Shard(0) DTensors over the expert mesh's
`ep_shard` dim (each rank materializes only its experts); MoE layers switch
to the RCCL all-to-all dispatch/combine handler. Router and shared expert
stay replicated.
"""

from torch import nn
from torch.distributed.device_mesh import DeviceMesh
from torch.distributed.tensor import Replicate, Shard

from ..module.block.moe import GroupedLinear, MoELayer
from ..module.block.moe.communications import RcclAllToAllCommunicationHandler
from .style import distribute_module_params


def parallelize_expert_parallel(
    module: nn.Module,
    mesh: DeviceMesh,
    shard_dim_name: str = "ep_shard",
) -> nn.Module:
    """Shard MoE experts over `shard_dim_name` of `mesh` (the expert domain).

    Must run BEFORE materialization: GroupedLinear weights are re-created with
    the local expert count so each rank only allocates its shard.
    """
    # Pipeline stages hold different modules: drop the pp dim (placements
    # over pp would make grad sync all-reduce across stages -- a deadlock).
    if mesh.mesh_dim_names and "pp" in mesh.mesh_dim_names and mesh.ndim > 1:
        keep = tuple(n for n in mesh.mesh_dim_names if n != "pp")
        mesh = mesh[keep]
    shard_dim = mesh.mesh_dim_names.index(shard_dim_name)
    ep_size = mesh.shape[shard_dim]
    ep_group = mesh.get_group(shard_dim)

    for sub in module.modules():
        if isinstance(sub, MoELayer):
            num_experts = sub.num_experts
            assert num_experts % ep_size == 0, (
                f"num_experts={num_experts} not divisible by ep={ep_size}"
            )
            local_experts = num_experts // ep_size
            for gl in sub.experts.modules():
                if isinstance(gl, GroupedLinear):
                    p = gl._parameters["weight"]
                    # Re-allocate with the LOCAL expert count on the same device.
                    local = p.data.new_empty(
                        (local_experts, p.shape[1], p.shape[2])
                    )
                    gl._parameters["weight"] = nn.Parameter(
                        local, requires_grad=p.requires_grad
                    )
                    from .style import shard_param_dim0

                    shard_param_dim0(gl, "weight", mesh, shard_dim)
            if ep_size > 1:
                sub.set_communication_handler(
                    RcclAllToAllCommunicationHandler(num_experts, ep_group)
                )
            # Router / shared expert / counters stay replicated over the
            # expert mesh (grad sync reduces over its Replicate dims).
            distribute_module_params(sub.router, mesh, recurse=True)
            if sub.shared_expert is not None:
                distribute_module_params(sub.shared_expert, mesh, recurse=True)
    return module
