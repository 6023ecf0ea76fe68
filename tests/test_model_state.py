"""model_state mapper DAG + streaming safetensors IO tests."""

import pytest
import torch
from torch import nn

from tests.helpers import run_distributed

from d9d_amd.model_state import (
    ChunkTensors,
    ConcatenateTensors,
    Identity,
    Parallel,
    PrefixScope,
    Rename,
    Sequential,
    Shard,
    StackTensors,
    Transpose,
    UnstackTensors,
    identity_mapper_from_module,
    load_model_state,
    read_model_state,
    save_module_state,
    write_model_state,
)


def test_leaf_mappers():
    t = torch.randn(3, 4)
    assert torch.equal(Rename("a", "b").apply({"a": t})["b"], t)
    assert torch.equal(
        Transpose("a").apply({"a": t})["a"], t.t().contiguous()
    )
    stack = StackTensors(["x0", "x1"], "x", dim=0)
    out = stack.apply({"x0": t, "x1": t + 1})
    assert out["x"].shape == (2, 3, 4)
    unstack = UnstackTensors("x", ["x0", "x1"], dim=0)
    back = unstack.apply(out)
    assert torch.equal(back["x0"], t) and torch.equal(back["x1"], t + 1)


def test_sequential_chains_groups():
    # rename then stack: net group {a.0, a.1} -> {stacked}
    m = Sequential(
        Parallel(Rename("a.0", "b.0"), Rename("a.1", "b.1")),
        StackTensors(["b.0", "b.1"], "stacked"),
    )
    groups = m.state_dependency_groups()
    assert len(groups) == 1
    g = groups[0]
    assert g.inputs == frozenset({"a.0", "a.1"})
    assert g.outputs == frozenset({"stacked"})
    t0, t1 = torch.randn(2), torch.randn(2)
    out = m.apply({"a.0": t0, "a.1": t1})
    assert torch.equal(out["stacked"], torch.stack([t0, t1]))


def test_sequential_gap_fill_passthrough():
    m = Sequential(
        Parallel(Identity("x"), Identity("y")),
        Rename("x", "z"),  # y passes through untouched
    )
    out = m.apply({"x": torch.ones(1), "y": torch.zeros(1)})
    assert set(out) == {"z", "y"}


def test_prefix_scope_and_shard():
    inner = Parallel(Rename("w", "weight"), Rename("b", "bias"))
    scoped = PrefixScope("layer0.", inner)
    out = scoped.apply({"layer0.w": torch.ones(1), "layer0.b": torch.zeros(1)})
    assert set(out) == {"layer0.weight", "layer0.bias"}

    sharded0 = Shard(inner, rank=0, world_size=2)
    sharded1 = Shard(inner, rank=1, world_size=2)
    g0 = sharded0.state_dependency_groups()
    g1 = sharded1.state_dependency_groups()
    assert len(g0) == 1 and len(g1) == 1 and g0 != g1


def test_write_read_roundtrip(tmp_path):
    state = {f"k{i}": torch.randn(16, 16) for i in range(5)}
    mapper = Parallel(*[Identity(k) for k in state])
    write_model_state(mapper, dict(state), tmp_path, shard_size_gb=1e-6)  # tiny shards
    files = list(tmp_path.glob("*.safetensors"))
    assert len(files) >= 2  # forced multi-shard
    loaded = dict(read_model_state(tmp_path))
    assert set(loaded) == set(state)
    for k in state:
        torch.testing.assert_close(loaded[k], state[k])


def test_module_save_load_roundtrip(tmp_path):
    torch.manual_seed(0)
    m1 = nn.Sequential(nn.Linear(4, 8), nn.Linear(8, 2))
    save_module_state(m1, tmp_path)
    m2 = nn.Sequential(nn.Linear(4, 8), nn.Linear(8, 2))
    loaded = load_model_state(m2, tmp_path, strict=True)
    assert len(loaded) == 4
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        torch.testing.assert_close(p1, p2)


def test_transform_on_write(tmp_path):
    # experts stored separately -> stacked on export (HF MODULE_LIST format flavor)
    experts = {f"expert.{i}.w": torch.randn(4, 4) for i in range(3)}
    mapper = StackTensors([f"expert.{i}.w" for i in range(3)], "experts.w", dim=0)
    write_model_state(mapper, dict(experts), tmp_path)
    loaded = dict(read_model_state(tmp_path))
    assert loaded["experts.w"].shape == (3, 4, 4)
    # and the reverse import path
    back = ChunkTensors("experts.w", [f"expert.{i}.w" for i in range(3)], dim=0)
    state = back.apply(loaded)
    for i in range(3):
        torch.testing.assert_close(
            state[f"expert.{i}.w"].squeeze(0), experts[f"expert.{i}.w"]
        )


def test_identity_mapper_from_module():
    m = nn.Linear(3, 3)
    mapper = identity_mapper_from_module(m)
    groups = mapper.state_dependency_groups()
    assert len(groups) == 2


def _distributed_export_ep(rank, world_size, tmp_dir):
    import torch
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor, Shard, distribute_tensor

    from d9d_amd.model_state.io import write_model_state_distributed
    from d9d_amd.model_state.mapper import GatherFullTensor, Parallel

    mesh = init_device_mesh("cpu", (world_size,), mesh_dim_names=("ep",))
    torch.manual_seed(1)
    full = torch.randn(4, 6)
    sharded = distribute_tensor(full, mesh, (Shard(0),))
    state = {"experts.weight": sharded, "norm.weight": torch.ones(6)}
    mapper = Parallel(*[GatherFullTensor(k) for k in state])

    # only rank 0 persists, but the gather inside is collective: every rank
    # must traverse (regression: this deadlocked / mismatched before)
    write_model_state_distributed(
        mapper, state, tmp_dir, is_writer=(rank == 0)
    )
    import torch.distributed as dist

    dist.barrier()
    if rank == 0:
        from safetensors.torch import load_file
        import json, pathlib

        index = json.loads((pathlib.Path(tmp_dir) / "model.safetensors.index.json").read_text())
        fname = index["weight_map"]["experts.weight"]
        data = load_file(str(pathlib.Path(tmp_dir) / fname))
        torch.testing.assert_close(data["experts.weight"], full)
    return True


@pytest.mark.distributed
def test_distributed_export_with_sharded_dtensor_ws2(tmp_path):
    import tempfile

    with tempfile.TemporaryDirectory() as d:
        assert all(run_distributed(_distributed_export_ep, world_size=2, args=(d,)))


def _distributed_save_load_roundtrip(rank, world_size, tmp_dir):
    """ws=2: export a model with EP-sharded DTensor experts, then stream the
    checkpoint back into a FRESH sharded model (auto-Distribute injection
    per DTensor param) and verify every rank's local shard."""
    import torch.distributed as dist
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor

    from d9d_amd.model_state import load_model_state, write_model_state_distributed
    from d9d_amd.model_state.mapper import Identity, Parallel

    mesh = init_device_mesh("cpu", (world_size,), mesh_dim_names=("ep_shard",))

    class _Experts(torch.nn.Module):
        def __init__(self, fill):
            super().__init__()
            local = torch.full((2, 4, 3), float(fill))
            placements = (torch.distributed.tensor.Shard(0),)
            self.weight = torch.nn.Parameter(
                DTensor.from_local(local, mesh, placements, run_check=False)
            )
            self.bias = torch.nn.Parameter(torch.arange(4.0))  # plain

    src = _Experts(fill=rank + 1)
    state = {
        "weight": src.weight.data.full_tensor(),
        "bias": src.bias.data,
    }
    write_model_state_distributed(
        Parallel(Identity("weight"), Identity("bias")), state, tmp_dir,
        is_writer=rank == 0,
    )
    dist.barrier()

    dst = _Experts(fill=0)
    loaded = load_model_state(dst, tmp_dir)
    assert set(loaded) == {"weight", "bias"}
    # every rank got ITS shard of the full (4, 4, 3) tensor
    torch.testing.assert_close(
        dst.weight.data.to_local(),
        torch.full((2, 4, 3), float(rank + 1)),
    )
    torch.testing.assert_close(dst.bias.data, torch.arange(4.0))
    return True


@pytest.mark.distributed
def test_distributed_save_load_roundtrip_ws2():
    import tempfile

    with tempfile.TemporaryDirectory() as d:
        assert all(run_distributed(_distributed_save_load_roundtrip, world_size=2, args=(d,)))


def _trainer_export_hf_reimport(rank, world_size, tmp_dir):
    """Trainer-level: train 1 step at DP2, export through the trainer, then
    re-load the export into a single-model instance on rank 0 and verify a
    forward pass matches across ranks' exports."""
    from d9d_amd.model_state import load_model_state
    from tests.test_loop import _build_trainer

    from d9d_amd.core.dist_context import DeviceMeshParameters
    from d9d_amd.parallel import parallelize_replicate

    def par(module, ctx):
        return parallelize_replicate(module, ctx.mesh_for("dense"))

    mesh = DeviceMeshParameters(data_parallel_replicate=2)
    trainer = _build_trainer(total_steps=1, parallelize=par, mesh=mesh)
    trainer.train()
    trainer.export(tmp_dir)

    import torch.distributed as dist

    dist.barrier()
    if rank == 0:
        from d9d_amd.module.model.qwen3_dense import (
            Qwen3DenseForCausalLM,
            Qwen3DenseModelParameters,
        )

        model = Qwen3DenseForCausalLM(Qwen3DenseModelParameters.tiny())
        loaded = load_model_state(model, tmp_dir)
        assert len(loaded) > 10
        ids = torch.randint(0, 100, (1, 8))
        out = model(input_ids=ids, labels=ids)
        assert torch.isfinite(out["loss"]).all()
    return True


@pytest.mark.distributed
def test_trainer_export_reimport_ws2():
    import tempfile

    with tempfile.TemporaryDirectory() as d:
        assert all(run_distributed(_trainer_export_hf_reimport, world_size=2, args=(d,)))


def test_select_child_leaf():
    """SelectChild picks one dim-0 slice into its own key (reference leaf)."""
    import torch

    from d9d_amd.model_state import SelectChild, Parallel

    t = torch.arange(24, dtype=torch.float32).view(3, 8)
    m = Parallel(*[SelectChild("experts", f"expert_{i}.w", i) for i in range(3)])
    out = m.apply({"experts": t})
    assert set(out) == {"expert_0.w", "expert_1.w", "expert_2.w"}
    for i in range(3):
        torch.testing.assert_close(out[f"expert_{i}.w"], t[i])
