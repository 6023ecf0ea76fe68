"""Single-rank RCCL smokes on a real GPU (world_size=1 process group):
pin the GPU-only code paths that multi-process CPU tests cannot reach —
the EP dispatch comm-stream/pinned-counts fast path and FSDP2-on-RCCL.
The 8-GPU scaling run composes these with real collectives."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture()
def world1_nccl():
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29611")
    created = False
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
        created = True
    torch.cuda.set_device(0)
    yield dist
    if created:
        dist.destroy_process_group()


def test_ep_dispatch_cuda_fast_path_world1(world1_nccl):
    """RcclAllToAllCommunicationHandler GPU path (comm stream, counts
    all-to-all, pinned D2H, cumsum regroup) with a self-group: the result
    must equal the local NoCommunicationHandler permute."""
    from d9d_amd.module.block.moe.communications import (
        NoCommunicationHandler,
        RcclAllToAllCommunicationHandler,
    )

    dist = world1_nccl
    torch.manual_seed(0)
    E, T, H, K = 8, 64, 32, 2
    tokens = torch.randn(T, H, dtype=torch.bfloat16, device="cuda")
    probs = torch.rand(T, K, dtype=torch.float32, device="cuda")
    indices = torch.randint(0, E, (T, K), device="cuda")

    rccl = RcclAllToAllCommunicationHandler(E, dist.group.WORLD)
    local = NoCommunicationHandler(E)

    rows_r, sizes_r, ctx_r = rccl.dispatch(tokens, probs, indices)
    rows_l, sizes_l, ctx_l = local.dispatch(tokens, probs, indices)
    torch.cuda.synchronize()

    assert sizes_r.cpu().tolist() == sizes_l.cpu().tolist()
    torch.testing.assert_close(rows_r, rows_l)

    expert_out = rows_r * 2.0
    back_r = rccl.combine(expert_out, ctx_r)
    back_l = local.combine(rows_l * 2.0, ctx_l)
    torch.cuda.synchronize()
    torch.testing.assert_close(back_r, back_l)


def test_ep_dispatch_cuda_autograd_world1(world1_nccl):
    """Gradients flow through the GPU dispatch/combine round trip."""
    from d9d_amd.module.block.moe.communications import (
        RcclAllToAllCommunicationHandler,
    )

    dist = world1_nccl
    torch.manual_seed(1)
    E, T, H, K = 4, 32, 16, 2
    tokens = torch.randn(T, H, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    probs = torch.rand(T, K, dtype=torch.float32, device="cuda", requires_grad=True)
    indices = torch.randint(0, E, (T, K), device="cuda")

    handler = RcclAllToAllCommunicationHandler(E, dist.group.WORLD)
    rows, sizes, ctx = handler.dispatch(tokens, probs, indices)
    out = handler.combine(rows * 3.0, ctx)
    out.sum().backward()
    torch.cuda.synchronize()
    assert tokens.grad is not None and torch.isfinite(tokens.grad.float()).all()
    assert probs.grad is not None and torch.isfinite(probs.grad).all()


def test_fsdp2_world1_smoke(world1_nccl):
    """fully_shard on RCCL (world 1): forward/backward/step with the forced
    SUM reduction overrides — pins the FSDP2-on-ROCm API/stream path
    (SURVEY risk #5)."""
    from torch import nn

    from d9d_amd.parallel import parallelize_fsdp

    dist = world1_nccl
    from torch.distributed.device_mesh import init_device_mesh

    mesh = init_device_mesh("cuda", (1,), mesh_dim_names=("dp_shard",))
    torch.manual_seed(2)
    model = nn.Sequential(
        nn.Linear(32, 64), nn.GELU(), nn.Linear(64, 32)
    ).cuda()
    parallelize_fsdp(model, mesh, shard_units=[model[0], model[2]])
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    x = torch.randn(8, 32, device="cuda")
    for _ in range(2):
        loss = model(x).pow(2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
