"""Sharding specs for PyTree microbatch/pipeline-state splitting.

Mirrors the reference's SpecReplicate/SpecShard contract
(d9d/core/sharding/spec.py:6-24). A spec tree has the same structure as the
data tree (or a single spec broadcast over all leaves).
"""

from dataclasses import dataclass
from typing import Union


@dataclass(frozen=True)
class SpecReplicate:
    """Leaf is replicated to every shard unchanged."""


@dataclass(frozen=True)
class SpecShard:
    """Leaf is split along `dim`.

    If `do_stack`, the leaf on the *unsharded* side carries an extra leading
    stack dimension: shard -> index along `dim`; unshard -> stack along `dim`.
    Otherwise chunks are concatenated / split with `tensor_split`.
    """

    dim: int = 0
    do_stack: bool = False


ShardingSpec = Union[SpecReplicate, SpecShard]
