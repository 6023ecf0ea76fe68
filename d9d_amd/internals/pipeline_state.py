"""Cross-microbatch scratch state (reference: d9d/internals/pipeline_state/).

Task callbacks write per-microbatch values (labels, losses, weights); reads
come back either per-shard or unsharded to the global view using the
auto-guessed spec (stack scalars, cat dim-0).
"""

from typing import Any

from ..core.sharding import guess_shard_spec, shard_tree, unshard_tree

STATE_LOSS = "loss"
STATE_LOSS_WEIGHT = "loss_weight"


class PipelineStateHandler:
    def __init__(self, num_microbatches: int) -> None:
        self.num_microbatches = num_microbatches
        self._global: dict[str, Any] = {}
        self._per_shard: dict[str, dict[int, Any]] = {}

    # -- global values sharded down -------------------------------------------

    def write_global(self, key: str, value: Any, spec=None) -> None:
        spec = spec if spec is not None else guess_shard_spec(value)
        shards = shard_tree(value, spec, self.num_microbatches)
        self._per_shard[key] = dict(enumerate(shards))
        self._global[key] = value

    # -- per-shard values merged up -------------------------------------------

    def write_shard(self, key: str, microbatch: int, value: Any) -> None:
        self._per_shard.setdefault(key, {})[microbatch] = value
        self._global.pop(key, None)

    def read_shard(self, key: str, microbatch: int) -> Any:
        return self._per_shard[key][microbatch]

    def read_global(self, key: str) -> Any:
        if key in self._global:
            return self._global[key]
        shards_map = self._per_shard[key]
        shards = [shards_map[i] for i in sorted(shards_map)]
        spec = guess_shard_spec(shards[0])
        merged = unshard_tree(shards, spec)
        self._global[key] = merged
        return merged

    def has(self, key: str) -> bool:
        return key in self._global or key in self._per_shard

    def microbatches_for(self, key: str) -> list[int]:
        return sorted(self._per_shard.get(key, {}))

    def reset(self) -> None:
        self._global.clear()
        self._per_shard.clear()
