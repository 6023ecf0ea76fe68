from .replicate import parallelize_replicate
from .fsdp import parallelize_fsdp, parallelize_hsdp
from .expert import parallelize_expert_parallel

__all__ = [
    "parallelize_replicate",
    "parallelize_fsdp",
    "parallelize_hsdp",
    "parallelize_expert_parallel",
]
