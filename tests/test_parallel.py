"""Distributed (gloo ws=2) tests: replicate + grad sync, EP dispatch, grad norm."""

import pytest
import torch
from torch import nn

from tests.helpers import run_distributed


def _replicate_grad_sync(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor

    from d9d_amd.internals.grad_sync import GradientSynchronizer
    from d9d_amd.parallel import parallelize_replicate

    torch.manual_seed(7)  # same init on both ranks
    model = nn.Sequential(nn.Linear(8, 16), nn.Linear(16, 4))
    ref = nn.Sequential(nn.Linear(8, 16), nn.Linear(16, 4))
    ref.load_state_dict(model.state_dict())

    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("dp",))
    parallelize_replicate(model, mesh)
    assert isinstance(model[0]._parameters["weight"], DTensor)
    # forward reads the local tensor
    assert not isinstance(model[0].weight, DTensor)

    sync = GradientSynchronizer(
        list(model.named_parameters()), accumulation_steps=1, bucket_bytes=1 << 20
    )

    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(4, 8)
    model(x).sum().backward()
    sync.wait()

    # reference: sum of both ranks' grads
    for r in range(2):
        torch.manual_seed(100 + r)
        xr = torch.randn(4, 8)
        ref(xr).sum().backward()

    for (name, p), (_, pr) in zip(model.named_parameters(), ref.named_parameters()):
        local_grad = p.grad.to_local() if isinstance(p.grad, DTensor) else p.grad
        torch.testing.assert_close(local_grad, pr.grad, rtol=1e-5, atol=1e-6)
    sync.remove()
    return True


@pytest.mark.distributed
def test_replicate_grad_sync_ws2():
    assert all(run_distributed(_replicate_grad_sync, world_size=2))


def _ep_dispatch_combine(rank, world_size):
    import torch.distributed as dist

    from d9d_amd.module.block.moe.communications import (
        RcclAllToAllCommunicationHandler,
    )

    num_experts = 4
    handler = RcclAllToAllCommunicationHandler(num_experts, dist.group.WORLD)

    torch.manual_seed(50 + rank)
    T, H, K = 6, 8, 2
    tokens = torch.randn(T, H)
    indices = torch.stack(
        [torch.randperm(num_experts)[:K] for _ in range(T)]
    )
    probs = torch.rand(T, K)
    probs = probs / probs.sum(-1, keepdim=True)

    rows, batch_sizes, ctx = handler.dispatch(tokens, probs, indices)
    assert batch_sizes.numel() == num_experts // world_size
    assert rows.shape[0] == int(batch_sizes.sum())

    # expert e multiplies by (e_global + 1); rank owns experts [rank*2, rank*2+2)
    out_rows = rows.clone()
    start = 0
    for e_local, n in enumerate(batch_sizes.tolist()):
        e_global = rank * (num_experts // world_size) + e_local
        out_rows[start : start + n] = rows[start : start + n] * (e_global + 1)
        start += n
    combined = handler.combine(out_rows, ctx)

    expected = torch.zeros_like(tokens)
    for t in range(T):
        for k in range(K):
            expected[t] += probs[t, k] * tokens[t] * (indices[t, k] + 1)
    torch.testing.assert_close(combined, expected, rtol=1e-4, atol=1e-5)
    return True


@pytest.mark.distributed
def test_ep_dispatch_combine_ws2():
    assert all(run_distributed(_ep_dispatch_combine, world_size=2))


def _ep_parallelize_moe_layer(rank, world_size):
    import torch.distributed as dist
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor

    from d9d_amd.module.block.moe import MoELayer
    from d9d_amd.parallel import parallelize_expert_parallel

    mesh = init_device_mesh(
        "cpu", (1, 1, 2, 1), mesh_dim_names=("pp", "ep_replicate", "ep_shard", "tp")
    )
    torch.manual_seed(3)
    layer = MoELayer(hidden_size=16, intermediate_size=8, num_experts=4, top_k=2)
    # Build the full-expert reference before sharding.
    layer.reset_parameters()
    full_weights = {
        n: p.detach().clone() for n, p in layer.experts.named_parameters()
    }
    ref_out = None
    x = None

    parallelize_expert_parallel(layer, mesh)
    # local shard: copy this rank's experts from the full weights
    with torch.no_grad():
        for n, p in layer.experts.named_parameters():
            full = full_weights[n.replace("._dt", "")]
            local = p.to_local() if isinstance(p, DTensor) else p
            e_local = local.shape[0]
            local.copy_(full[rank * e_local : (rank + 1) * e_local])

    torch.manual_seed(80 + rank)
    x = torch.randn(2, 5, 16)

    out = layer(x)

    # reference: same layer logic with all experts local
    ref_layer = MoELayer(hidden_size=16, intermediate_size=8, num_experts=4, top_k=2)
    ref_layer.reset_parameters()
    with torch.no_grad():
        for n, p in ref_layer.experts.named_parameters():
            p.copy_(full_weights[n])
        ref_layer.router.gate.weight.copy_(
            layer.router.gate.weight
            if not isinstance(layer.router._parameters.get("weight", None), DTensor)
            else layer.router.gate.weight
        )
    ref_out = ref_layer(x)
    torch.testing.assert_close(out, ref_out, rtol=1e-4, atol=1e-5)
    return True


@pytest.mark.distributed
def test_ep_moe_layer_matches_local_ws2():
    assert all(run_distributed(_ep_parallelize_moe_layer, world_size=2))


def _grad_norm_clip(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh

    from d9d_amd.internals.grad_norm import clip_grad_norm_distributed_
    from d9d_amd.parallel import parallelize_replicate

    torch.manual_seed(5)
    model = nn.Linear(4, 4)
    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("dp",))
    parallelize_replicate(model, mesh)
    # identical grads on both ranks (replicated params => replicated grads)
    for p in model.parameters():
        local = p.to_local()
        g = torch.ones_like(local)
        from torch.distributed.tensor import DTensor

        p.grad = DTensor.from_local(g, mesh, p.placements, run_check=False)
    norm = clip_grad_norm_distributed_(list(model.parameters()), max_norm=1.0)
    expected = torch.ones(20).norm()  # 16 weights + 4 bias, replicated (not sharded)
    torch.testing.assert_close(norm, expected, rtol=1e-5, atol=1e-6)
    for p in model.parameters():
        local = p.grad.to_local()
        assert local.abs().max() <= 1.0 / expected + 1e-4
    return True


@pytest.mark.distributed
def test_grad_norm_clip_ws2():
    assert all(run_distributed(_grad_norm_clip, world_size=2))


def _vocab_parallel_cce(rank, world_size):
    import torch.distributed as dist

    from d9d_amd.ops.cce import VocabParallelOptions, linear_cross_entropy

    torch.manual_seed(2)
    T, H, V = 12, 16, 40
    e = torch.randn(T, H)
    c = torch.randn(V, H) * 0.1
    targets = torch.randint(0, V, (T,))
    targets[0] = -100

    # reference: full-vocab loss
    e_ref = e.clone().requires_grad_(True)
    c_ref = c.clone().requires_grad_(True)
    ref = linear_cross_entropy(e_ref, c_ref, targets)
    ref.sum().backward()

    # vocab-parallel: this rank owns rows [rank*20, rank*20+20)
    shard = 20
    e_vp = e.clone().requires_grad_(True)
    c_local = c[rank * shard : (rank + 1) * shard].clone().requires_grad_(True)
    vp = VocabParallelOptions(dist.group.WORLD, rank * shard, (rank + 1) * shard)
    loss = linear_cross_entropy(e_vp, c_local, targets, vocab_parallel=vp)
    loss.sum().backward()

    torch.testing.assert_close(loss, ref, rtol=1e-4, atol=1e-5)
    # e grads: backward all-reduces the per-shard partials over the vp group
    # (reference reduce_e_grad), so every rank holds the full de.
    torch.testing.assert_close(e_vp.grad, e_ref.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(
        c_local.grad, c_ref.grad[rank * shard : (rank + 1) * shard], rtol=1e-4, atol=1e-5
    )
    return True


@pytest.mark.distributed
def test_vocab_parallel_cce_ws2():
    assert all(run_distributed(_vocab_parallel_cce, world_size=2))


def _fsdp_grad_matches_replicated(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor

    from d9d_amd.parallel import parallelize_fsdp

    torch.manual_seed(21)
    model = nn.Sequential(nn.Linear(16, 32), nn.Tanh(), nn.Linear(32, 8))
    ref = nn.Sequential(nn.Linear(16, 32), nn.Tanh(), nn.Linear(32, 8))
    ref.load_state_dict(model.state_dict())

    mesh = init_device_mesh("cpu", (world_size,), mesh_dim_names=("dp_shard",))
    parallelize_fsdp(model, mesh, shard_units=[model[0], model[2]])
    # params are sharded DTensors now
    assert isinstance(next(model.parameters()), DTensor)

    torch.manual_seed(300 + rank)
    x = torch.randn(6, 16)
    model(x).pow(2).sum().backward()

    # reference: sum of both ranks' grads (sum reduction, divide factor 1)
    for r in range(world_size):
        torch.manual_seed(300 + r)
        xr = torch.randn(6, 16)
        ref(xr).pow(2).sum().backward()

    for (n, p), (_, pr) in zip(model.named_parameters(), ref.named_parameters()):
        assert p.grad is not None, n
        full = p.grad.full_tensor() if isinstance(p.grad, DTensor) else p.grad
        torch.testing.assert_close(full, pr.grad, rtol=1e-4, atol=1e-5)
    return True


@pytest.mark.distributed
def test_fsdp_grad_sum_ws2():
    assert all(run_distributed(_fsdp_grad_matches_replicated, world_size=2))


def _hsdp_grad_matches(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor

    from d9d_amd.parallel import parallelize_hsdp

    torch.manual_seed(22)
    model = nn.Sequential(nn.Linear(16, 32), nn.Tanh(), nn.Linear(32, 8))
    ref = nn.Sequential(nn.Linear(16, 32), nn.Tanh(), nn.Linear(32, 8))
    ref.load_state_dict(model.state_dict())

    mesh = init_device_mesh(
        "cpu", (2, 2), mesh_dim_names=("dp_replicate", "dp_shard")
    )
    parallelize_hsdp(model, mesh, shard_units=[model[0], model[2]])
    assert isinstance(next(model.parameters()), DTensor)

    torch.manual_seed(400 + rank)
    x = torch.randn(6, 16)
    model(x).pow(2).sum().backward()

    for r in range(world_size):
        torch.manual_seed(400 + r)
        xr = torch.randn(6, 16)
        ref(xr).pow(2).sum().backward()

    for (n, p), (_, pr) in zip(model.named_parameters(), ref.named_parameters()):
        full = p.grad.full_tensor() if isinstance(p.grad, DTensor) else p.grad
        torch.testing.assert_close(full, pr.grad, rtol=1e-4, atol=1e-5)
    return True


@pytest.mark.distributed
def test_hsdp_grad_sum_ws4():
    assert all(run_distributed(_hsdp_grad_matches, world_size=4))


def _replicate_broadcast(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor

    from d9d_amd.parallel import parallelize_replicate

    torch.manual_seed(900 + rank)  # deliberately DIFFERENT init per rank
    model = nn.Linear(4, 4)
    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("dp",))
    parallelize_replicate(model, mesh)
    import torch.distributed as dist

    w = model._parameters["weight"]
    local = w.to_local() if isinstance(w, DTensor) else w
    gathered = [torch.empty_like(local) for _ in range(world_size)]
    dist.all_gather(gathered, local)
    assert torch.equal(gathered[0], gathered[1]), "replicas must agree after broadcast"
    return True


@pytest.mark.distributed
def test_replicate_broadcasts_initial_weights_ws2():
    assert all(run_distributed(_replicate_broadcast, world_size=2))
