from .params import DeviceMeshParameters
from .configured import DistributedContext
from .domains import (
    MeshDomain,
    REGULAR_DOMAIN,
    DENSE_DOMAIN,
    EXPERT_DOMAIN,
    BATCH_DOMAIN,
    FLAT_DOMAIN,
)

__all__ = [
    "DeviceMeshParameters",
    "DistributedContext",
    "MeshDomain",
    "REGULAR_DOMAIN",
    "DENSE_DOMAIN",
    "EXPERT_DOMAIN",
    "BATCH_DOMAIN",
    "FLAT_DOMAIN",
]
