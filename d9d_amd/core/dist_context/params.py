"""Topology parameters — the single source of truth for the 7-degree mesh.

Mirrors the reference contract (d9d/core/dist_context/params.py:9-97):
seven degrees, EP-divisibility validation, and a `build()` that initializes
process groups and returns a `DistributedContext`.
"""

from dataclasses import dataclass, field

from .domains import (
    ALL_DOMAINS,
    BATCH_DOMAIN,
    DENSE_DOMAIN,
    EXPERT_DOMAIN,
    FLAT_DOMAIN,
    REGULAR_DOMAIN,
)


@dataclass(frozen=True)
class DeviceMeshParameters:
    """Degrees of every parallelism axis.

    world = pp * dp_replicate * dp_shard * cp_shard * cp_replicate * tp.
    ep refactors the dp×cp block: ep_shard = ep, ep_replicate = dp*cp / ep.
    """

    pipeline_parallel: int = 1
    data_parallel_replicate: int = 1
    data_parallel_shard: int = 1
    context_parallel_shard: int = 1
    context_parallel_replicate: int = 1
    tensor_parallel: int = 1
    expert_parallel: int = 1

    # Collective timeouts (seconds); the loop's TimeoutManager swaps between them.
    init_timeout_seconds: float = field(default=600.0, compare=False)
    step_timeout_seconds: float = field(default=120.0, compare=False)

    def __post_init__(self) -> None:
        for name, deg in self.degrees().items():
            if deg < 1:
                raise ValueError(f"mesh degree {name!r} must be >= 1, got {deg}")
        dp_cp = (
            self.data_parallel_replicate
            * self.data_parallel_shard
            * self.context_parallel_shard
            * self.context_parallel_replicate
        )
        if dp_cp % self.expert_parallel != 0:
            raise ValueError(
                "expert_parallel must divide dp_replicate*dp_shard*cp_shard*cp_replicate "
                f"(= {dp_cp}), got expert_parallel={self.expert_parallel}"
            )

    def degrees(self) -> dict[str, int]:
        return {
            "pp": self.pipeline_parallel,
            "dp_replicate": self.data_parallel_replicate,
            "dp_shard": self.data_parallel_shard,
            "cp_shard": self.context_parallel_shard,
            "cp_replicate": self.context_parallel_replicate,
            "tp": self.tensor_parallel,
            "ep": self.expert_parallel,
        }

    @property
    def world_size(self) -> int:
        return (
            self.pipeline_parallel
            * self.data_parallel_replicate
            * self.data_parallel_shard
            * self.context_parallel_shard
            * self.context_parallel_replicate
            * self.tensor_parallel
        )

    def domain_degrees(self) -> dict[str, int]:
        """Degrees for every dim name used by any domain."""
        d = self.degrees()
        dp = d["dp_replicate"] * d["dp_shard"]
        cp = d["cp_shard"] * d["cp_replicate"]
        ep_shard = d["ep"]
        return {
            **{k: v for k, v in d.items() if k != "ep"},
            "dp_cp_shard": d["dp_shard"] * d["cp_shard"],
            "dp": dp,
            "cp": cp,
            "ep_shard": ep_shard,
            "ep_replicate": dp * cp // ep_shard,
            "world": self.world_size,
        }

    def domain_shapes(self) -> dict[str, tuple[tuple[str, ...], tuple[int, ...]]]:
        deg = self.domain_degrees()
        return {
            dom.name: (dom.dim_names, dom.shape_from(deg))
            for dom in ALL_DOMAINS
        }

    def build(self, device_type: str | None = None) -> "DistributedContext":
        from .configured import DistributedContext

        return DistributedContext.create(self, device_type=device_type)


__all__ = [
    "DeviceMeshParameters",
    "REGULAR_DOMAIN",
    "DENSE_DOMAIN",
    "EXPERT_DOMAIN",
    "BATCH_DOMAIN",
    "FLAT_DOMAIN",
]
