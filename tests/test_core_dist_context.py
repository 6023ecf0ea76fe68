import pytest
import torch

from d9d_amd.core.dist_context import DeviceMeshParameters
from tests.helpers import run_distributed


def test_world_size_math():
    p = DeviceMeshParameters(
        pipeline_parallel=4,
        data_parallel_replicate=2,
        data_parallel_shard=1,
        expert_parallel=2,
    )
    assert p.world_size == 8
    deg = p.domain_degrees()
    assert deg["dp"] == 2 and deg["cp"] == 1
    assert deg["ep_shard"] == 2 and deg["ep_replicate"] == 1


def test_ep_divisibility_validation():
    with pytest.raises(ValueError):
        DeviceMeshParameters(data_parallel_replicate=2, expert_parallel=3)


def test_degree_positive_validation():
    with pytest.raises(ValueError):
        DeviceMeshParameters(pipeline_parallel=0)


def test_domain_shapes_consistent():
    p = DeviceMeshParameters(
        pipeline_parallel=2,
        data_parallel_replicate=2,
        data_parallel_shard=2,
        context_parallel_shard=1,
        tensor_parallel=1,
        expert_parallel=4,
    )
    shapes = p.domain_shapes()
    import math

    for name, (_, shape) in shapes.items():
        assert math.prod(shape) == p.world_size, name


def test_local_context():
    ctx = DeviceMeshParameters().build(device_type="cpu")
    assert ctx.world_size == 1
    assert not ctx.is_distributed
    assert ctx.is_main_process
    ctx.wait_world()  # no-op
    with pytest.raises(KeyError):
        ctx.mesh_for("dense")


def _build_ctx_ws2(rank, world_size):
    p = DeviceMeshParameters(data_parallel_replicate=2)
    ctx = p.build(device_type="cpu")
    mesh = ctx.mesh_for("dense")
    assert mesh.shape == (1, 2, 1, 1, 1)
    batch = ctx.mesh_for("batch")
    dp_rank = batch.get_local_rank("dp")
    ctx.wait_world()
    return dp_rank


@pytest.mark.distributed
def test_mesh_build_gloo_ws2():
    results = run_distributed(_build_ctx_ws2, world_size=2)
    assert sorted(results) == [0, 1]


def _all_gather_variadic(rank, world_size):
    from d9d_amd.core.dist_ops import all_gather_variadic_shape

    t = torch.full((rank + 1, 3), float(rank))
    out = all_gather_variadic_shape(t)
    assert [tuple(x.shape) for x in out] == [(1, 3), (2, 3)]
    assert out[1][0, 0].item() == 1.0
    return True


@pytest.mark.distributed
def test_all_gather_variadic_shape_ws2():
    assert all(run_distributed(_all_gather_variadic, world_size=2))


def _gather_variadic(rank, world_size):
    from d9d_amd.core.dist_ops import gather_variadic_shape

    t = torch.full((2 * rank + 1,), float(rank))
    out = gather_variadic_shape(t, dst=0)
    if rank == 0:
        assert [tuple(x.shape) for x in out] == [(1,), (3,)]
        assert out[1][0].item() == 1.0
    else:
        assert out is None
    return True


@pytest.mark.distributed
def test_gather_variadic_shape_ws2():
    assert all(run_distributed(_gather_variadic, world_size=2))
