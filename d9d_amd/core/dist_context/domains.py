"""Named mesh domains — different factorizations of the same world.

Mirrors the reference's five device-mesh domains
(reference: d9d/core/dist_context/device_mesh_domains.py:39-171) but built
MI355X-first: every domain is one `init_device_mesh` call whose process groups
become RCCL communicators over xGMI.

A *domain* is a named view of the world as a dense grid. All domains must
factorize the world consistently: the rank→coordinate map of every domain is a
fusion of adjacent dims of the canonical ordering

    (pp, dp_replicate, dp_shard, cp_shard, cp_replicate, tp)

(outermost → innermost), so collectives on a fused dim of one domain address
exactly the union of the corresponding dims in another.
"""

from dataclasses import dataclass


@dataclass(frozen=True)
class MeshDomain:
    """A named factorization of the world into mesh dims."""

    name: str
    dim_names: tuple[str, ...]

    def shape_from(self, degrees: dict[str, int]) -> tuple[int, ...]:
        return tuple(degrees[d] for d in self.dim_names)


# Full-resolution domain: one dim per parallelism degree.
REGULAR_DOMAIN = MeshDomain(
    name="regular",
    dim_names=("pp", "dp_replicate", "dp_shard", "cp_shard", "cp_replicate", "tp"),
)

# FSDP-facing domain: dp_shard and cp_shard fused — parameters are sharded over
# both at once (reference: device_mesh_domains.py:99-121).
DENSE_DOMAIN = MeshDomain(
    name="dense",
    dim_names=("pp", "dp_replicate", "dp_cp_shard", "cp_replicate", "tp"),
)

# Expert-parallel domain: the dp×cp block refactored into
# (ep_replicate, ep_shard); experts are Shard(0) over ep_shard
# (reference: device_mesh_domains.py:69-93). tp is kept as an explicit inner
# dim so TP can compose with EP (the reference declares tp but rejects tp>1).
EXPERT_DOMAIN = MeshDomain(
    name="expert",
    dim_names=("pp", "ep_replicate", "ep_shard", "tp"),
)

# Batch-sharding domain: how input batches divide (reference: 127-147).
BATCH_DOMAIN = MeshDomain(
    name="batch",
    dim_names=("pp", "dp", "cp", "tp"),
)

# Whole-world domain (reference: 153-171).
FLAT_DOMAIN = MeshDomain(
    name="flat",
    dim_names=("world",),
)

ALL_DOMAINS = (REGULAR_DOMAIN, DENSE_DOMAIN, EXPERT_DOMAIN, BATCH_DOMAIN, FLAT_DOMAIN)
