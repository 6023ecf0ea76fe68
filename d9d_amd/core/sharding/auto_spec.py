"""Auto-spec helpers (reference: d9d/core/sharding/auto_spec.py:27,52).

`guess_shard_spec` mirrors the pipeline-state storage heuristic
(d9d/internals/pipeline_state/storage.py): scalars stack, tensors cat on dim 0.
"""

from typing import Any

import torch
import torch.utils._pytree as pytree

from .spec import SpecReplicate, SpecShard


def shard_spec_on_dim(tree: Any, dim: int = 0) -> Any:
    """Spec tree sharding every tensor leaf on `dim`, replicating the rest."""
    return pytree.tree_map(
        lambda leaf: SpecShard(dim=dim)
        if isinstance(leaf, (torch.Tensor, list, tuple))
        else SpecReplicate(),
        tree,
    )


def shard_spec_nothing(tree: Any) -> Any:
    return pytree.tree_map(lambda _: SpecReplicate(), tree)


def guess_shard_spec(tree: Any) -> Any:
    """Stack 0-dim tensors, cat >=1-dim tensors on dim 0, replicate non-tensors."""

    def leaf_spec(leaf: Any) -> Any:
        if isinstance(leaf, torch.Tensor):
            if leaf.dim() == 0:
                return SpecShard(dim=0, do_stack=True)
            return SpecShard(dim=0)
        return SpecReplicate()

    return pytree.tree_map(leaf_spec, tree)
