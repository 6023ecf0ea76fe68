from .spec import SpecReplicate, SpecShard, ShardingSpec
from .shard import shard_tree
from .unshard import unshard_tree
from .auto_spec import shard_spec_on_dim, shard_spec_nothing, guess_shard_spec

__all__ = [
    "SpecReplicate",
    "SpecShard",
    "ShardingSpec",
    "shard_tree",
    "unshard_tree",
    "shard_spec_on_dim",
    "shard_spec_nothing",
    "guess_shard_spec",
]
