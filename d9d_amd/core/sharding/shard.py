"""shard_tree — split a PyTree of tensors/lists into N shards.

Reference contract: d9d/core/sharding/shard.py:105 (tensor_split semantics,
list splitting, spec broadcast). A spec node marks a leaf of the data tree, so
list-valued leaves (sharded by element count) are supported.
"""

from typing import Any

import torch

from .spec import ShardingSpec, SpecReplicate, SpecShard


def _is_spec(x: Any) -> bool:
    return isinstance(x, (SpecReplicate, SpecShard))


def _shard_leaf(leaf: Any, spec: ShardingSpec, num_shards: int) -> list[Any]:
    if isinstance(spec, SpecReplicate):
        return [leaf] * num_shards

    assert isinstance(spec, SpecShard)
    if isinstance(leaf, torch.Tensor):
        if spec.do_stack:
            if leaf.size(spec.dim) != num_shards:
                raise ValueError(
                    f"stacked shard dim {spec.dim} has size {leaf.size(spec.dim)}, "
                    f"expected {num_shards}"
                )
            return [t.contiguous() for t in leaf.unbind(spec.dim)]
        return [t.contiguous() for t in leaf.tensor_split(num_shards, dim=spec.dim)]
    if isinstance(leaf, (list, tuple)):
        n = len(leaf)
        base, rem = divmod(n, num_shards)
        shards = []
        start = 0
        for i in range(num_shards):
            size = base + (1 if i < rem else 0)
            shards.append(type(leaf)(leaf[start : start + size]))
            start += size
        return shards
    raise TypeError(f"cannot shard leaf of type {type(leaf)!r}")


def _shard_node(node: Any, spec: Any, num_shards: int) -> list[Any]:
    """Co-traverse data and spec; a spec instance marks a data leaf.

    A single spec broadcast over a dict descends; tensors and sequences under
    a spec are leaves (sequences shard by element count).
    """
    if _is_spec(spec):
        if isinstance(node, dict):
            per_key = {k: _shard_node(node[k], spec, num_shards) for k in node}
            return [{k: per_key[k][i] for k in node} for i in range(num_shards)]
        return _shard_leaf(node, spec, num_shards)
    if isinstance(spec, dict):
        if not isinstance(node, dict) or set(node) != set(spec):
            raise ValueError(f"spec/data dict mismatch: {set(spec)} vs data {type(node)}")
        per_key = {k: _shard_node(node[k], spec[k], num_shards) for k in node}
        return [{k: per_key[k][i] for k in node} for i in range(num_shards)]
    if isinstance(spec, (list, tuple)):
        if not isinstance(node, (list, tuple)) or len(node) != len(spec):
            raise ValueError("spec/data sequence mismatch")
        per_elem = [_shard_node(n, s, num_shards) for n, s in zip(node, spec)]
        return [type(node)(pe[i] for pe in per_elem) for i in range(num_shards)]
    raise TypeError(f"invalid spec node of type {type(spec)!r}")


def shard_tree(tree: Any, spec_tree: Any, num_shards: int) -> list[Any]:
    """Split `tree` into `num_shards` trees of the same structure.

    `spec_tree` is either a single spec (broadcast over all tensor leaves via
    structural descent) or a tree whose spec instances mark the data leaves.
    """
    return _shard_node(tree, spec_tree, num_shards)
