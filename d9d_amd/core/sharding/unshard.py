"""unshard_tree — inverse of shard_tree (reference: d9d/core/sharding/unshard.py:60)."""

from typing import Any

import torch

from .spec import ShardingSpec, SpecReplicate, SpecShard


def _is_spec(x: Any) -> bool:
    return isinstance(x, (SpecReplicate, SpecShard))


def _unshard_leaf(shards: list[Any], spec: ShardingSpec) -> Any:
    if isinstance(spec, SpecReplicate):
        return shards[0]

    assert isinstance(spec, SpecShard)
    first = shards[0]
    if isinstance(first, torch.Tensor):
        if spec.do_stack:
            return torch.stack(shards, dim=spec.dim)
        return torch.cat(shards, dim=spec.dim)
    if isinstance(first, (list, tuple)):
        out: list[Any] = []
        for s in shards:
            out.extend(s)
        return type(first)(out)
    raise TypeError(f"cannot unshard leaf of type {type(first)!r}")


def _unshard_node(shards: list[Any], spec: Any) -> Any:
    first = shards[0]
    if _is_spec(spec):
        if isinstance(first, dict):
            return {k: _unshard_node([s[k] for s in shards], spec) for k in first}
        return _unshard_leaf(shards, spec)
    if isinstance(spec, dict):
        return {k: _unshard_node([s[k] for s in shards], spec[k]) for k in first}
    if isinstance(spec, (list, tuple)):
        return type(first)(
            _unshard_node([s[i] for s in shards], spec[i]) for i in range(len(first))
        )
    raise TypeError(f"invalid spec node of type {type(spec)!r}")


def unshard_tree(shards: list[Any], spec_tree: Any) -> Any:
    """Merge same-structure shard trees back into one tree."""
    if not shards:
        raise ValueError("no shards to unshard")
    return _unshard_node(shards, spec_tree)
