from .context import parallelize_context_parallel, shard_sequence
from .expert import parallelize_expert_parallel
from .fsdp import parallelize_fsdp, parallelize_hsdp
from .replicate import parallelize_replicate
from .tensor import parallelize_tensor_parallel

__all__ = [
    "parallelize_replicate",
    "parallelize_fsdp",
    "parallelize_hsdp",
    "parallelize_expert_parallel",
    "parallelize_tensor_parallel",
    "parallelize_context_parallel",
    "shard_sequence",
]
