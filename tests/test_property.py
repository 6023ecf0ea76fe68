"""Property-based round-trips (hypothesis) for sharding and state mappers."""

import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from d9d_amd.core.sharding import (
    SpecReplicate,
    SpecShard,
    shard_tree,
    unshard_tree,
)
from d9d_amd.model_state.mapper import (
    ChunkTensors,
    ConcatenateTensors,
    SliceRows,
    StackTensors,
    UnstackTensors,
)


@settings(max_examples=50, deadline=None, derandomize=True)
@given(
    rows=st.integers(min_value=1, max_value=33),
    cols=st.integers(min_value=1, max_value=8),
    n=st.integers(min_value=1, max_value=5),
)
def test_shard_unshard_roundtrip(rows, cols, n):
    tree = {
        "a": torch.randn(rows * n, cols),
        "nested": {"b": torch.randn(rows * n, 3)},
        "scalar_list": [torch.randn(2) for _ in range(n)],
    }
    spec = {
        "a": SpecShard(0),
        "nested": {"b": SpecShard(0)},
        "scalar_list": SpecShard(0),
    }
    shards = shard_tree(tree, spec, n)
    assert len(shards) == n
    merged = unshard_tree(shards, spec)
    torch.testing.assert_close(merged["a"], tree["a"])
    torch.testing.assert_close(merged["nested"]["b"], tree["nested"]["b"])
    for got, want in zip(merged["scalar_list"], tree["scalar_list"]):
        torch.testing.assert_close(got, want)


@settings(max_examples=50, deadline=None, derandomize=True)
@given(
    sizes=st.lists(st.integers(min_value=1, max_value=9), min_size=1, max_size=5),
    cols=st.integers(min_value=1, max_value=6),
)
def test_slice_concat_roundtrip(sizes, cols):
    full = torch.randn(sum(sizes), cols)
    names = [(f"p{i}", s) for i, s in enumerate(sizes)]
    parts = SliceRows("x", names).apply({"x": full})
    back = ConcatenateTensors([n for n, _ in names], "x").apply(parts)
    torch.testing.assert_close(back["x"], full)


@settings(max_examples=50, deadline=None, derandomize=True)
@given(
    n=st.integers(min_value=1, max_value=6),
    shape=st.tuples(
        st.integers(min_value=1, max_value=5), st.integers(min_value=1, max_value=5)
    ),
)
def test_stack_unstack_roundtrip(n, shape):
    tensors = {f"t{i}": torch.randn(*shape) for i in range(n)}
    stacked = StackTensors(list(tensors), "s").apply(dict(tensors))
    assert stacked["s"].shape == (n, *shape)
    back = UnstackTensors("s", list(tensors)).apply(stacked)
    for k, v in tensors.items():
        torch.testing.assert_close(back[k], v)


@settings(max_examples=30, deadline=None, derandomize=True)
@given(
    n=st.integers(min_value=1, max_value=4),
    rows_per=st.integers(min_value=1, max_value=6),
)
def test_chunk_concat_roundtrip(n, rows_per):
    full = torch.randn(n * rows_per, 3)
    names = [f"c{i}" for i in range(n)]
    parts = ChunkTensors("x", names).apply({"x": full})
    back = ConcatenateTensors(names, "x").apply(parts)
    torch.testing.assert_close(back["x"], full)
