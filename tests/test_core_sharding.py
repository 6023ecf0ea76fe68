import torch

from d9d_amd.core.sharding import (
    SpecReplicate,
    SpecShard,
    guess_shard_spec,
    shard_tree,
    unshard_tree,
)


def test_shard_tree_roundtrip_tensors():
    tree = {"x": torch.arange(24).reshape(8, 3), "y": torch.randn(4, 6)}
    spec = {"x": SpecShard(dim=0), "y": SpecShard(dim=1)}
    shards = shard_tree(tree, spec, 2)
    assert len(shards) == 2
    assert shards[0]["x"].shape == (4, 3)
    assert shards[0]["y"].shape == (4, 3)
    merged = unshard_tree(shards, spec)
    assert torch.equal(merged["x"], tree["x"])
    assert torch.equal(merged["y"], tree["y"])


def test_shard_tree_uneven_split():
    t = torch.arange(10)
    shards = shard_tree(t, SpecShard(dim=0), 4)
    sizes = [s.numel() for s in shards]
    assert sum(sizes) == 10
    assert max(sizes) - min(sizes) <= 1
    assert torch.equal(unshard_tree(shards, SpecShard(dim=0)), t)


def test_shard_tree_replicate_and_lists():
    tree = {"ids": [1, 2, 3, 4, 5], "flag": "keep"}
    spec = {"ids": SpecShard(dim=0), "flag": SpecReplicate()}
    shards = shard_tree(tree, spec, 2)
    assert shards[0]["flag"] == "keep" and shards[1]["flag"] == "keep"
    assert shards[0]["ids"] == [1, 2, 3]
    assert shards[1]["ids"] == [4, 5]


def test_stack_spec_roundtrip():
    t = torch.randn(3, 5)
    shards = shard_tree(t, SpecShard(dim=0, do_stack=True), 3)
    assert shards[0].shape == (5,)
    merged = unshard_tree(shards, SpecShard(dim=0, do_stack=True))
    assert torch.equal(merged, t)


def test_guess_shard_spec():
    tree = {"loss": torch.tensor(1.0), "h": torch.randn(4, 2), "s": "x"}
    spec = guess_shard_spec(tree)
    assert spec["loss"] == SpecShard(dim=0, do_stack=True)
    assert spec["h"] == SpecShard(dim=0)
    assert spec["s"] == SpecReplicate()
