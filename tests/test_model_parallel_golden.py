"""Global-vs-parallel golden harness (reference test pattern #1:
test/d9d_test/modules/model/sequence/causal_lm/test_distributed.py — build
the single-process model, build the parallelized copy with weights copied
over, run the same data, compare loss and per-parameter gradient ANGLE +
norm distances with explicit tolerances)."""

import pytest
import torch

from tests.helpers import run_distributed


def grad_angle(a: torch.Tensor, b: torch.Tensor) -> float:
    """Angle (radians) between flattened gradients."""
    af, bf = a.float().flatten(), b.float().flatten()
    cos = torch.dot(af, bf) / (af.norm() * bf.norm()).clamp_min(1e-20)
    return float(torch.arccos(cos.clamp(-1.0, 1.0)))


def grad_norm_ratio(a: torch.Tensor, b: torch.Tensor) -> float:
    """|‖a‖ - ‖b‖| / ‖b‖."""
    na, nb = a.float().norm(), b.float().norm()
    return float((na - nb).abs() / nb.clamp_min(1e-20))


def assert_grads_close(got: dict, ref: dict, max_angle=5e-3, max_norm=5e-3):
    missing = set(ref) - set(got)
    assert not missing, f"missing grads: {sorted(missing)[:5]}"
    for name, g in got.items():
        r = ref[name]
        ang = grad_angle(g, r)
        nrm = grad_norm_ratio(g, r)
        assert ang <= max_angle, f"{name}: grad angle {ang:.2e}"
        assert nrm <= max_norm, f"{name}: grad norm ratio {nrm:.2e}"


def _tp2_whole_model(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor

    from d9d_amd.module.model.qwen3_dense import (
        Qwen3DenseForCausalLM,
        Qwen3DenseModelParameters,
    )
    from d9d_amd.parallel import parallelize_tensor_parallel

    p = Qwen3DenseModelParameters(
        hidden_size=64,
        intermediate_size=128,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=16,
        num_hidden_layers=2,
        split_vocab_size={"regular": 120, "special": 8},
    )
    torch.manual_seed(77)
    global_model = Qwen3DenseForCausalLM(p)
    global_model.init_weights()

    torch.manual_seed(77)
    par_model = Qwen3DenseForCausalLM(p)
    par_model.init_weights()
    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("tp",))
    parallelize_tensor_parallel(par_model, mesh)
    # copy global weights into the TP shards
    with torch.no_grad():
        gsd = global_model.state_dict()
        for name, param in par_model.named_parameters():
            src = gsd[name]
            if isinstance(param, DTensor):
                local = param.to_local()
                placement = param.placements[0]
                if placement.is_shard():
                    dim = placement.dim
                    size = local.shape[dim]
                    local.copy_(src.narrow(dim, rank * size, size))
                else:
                    local.copy_(src)
            elif param.shape != src.shape:
                # plain local shard (vocab-parallel LM head): rows
                # [rank*sh, (rank+1)*sh) of the segment
                sh = param.shape[0]
                param.copy_(src.narrow(0, rank * sh, sh))
            else:
                param.copy_(src)

    torch.manual_seed(5)  # same data on both ranks (TP replicates data)
    ids = torch.randint(0, p.vocab_size, (2, 24))
    out_g = global_model(input_ids=ids, labels=ids)
    out_p = par_model(input_ids=ids, labels=ids)
    loss_g = out_g["loss"].mean()
    loss_p = out_p["loss"].mean()
    torch.testing.assert_close(loss_p, loss_g, rtol=1e-4, atol=1e-5)
    loss_g.backward()
    loss_p.backward()

    ref = {n: q.grad for n, q in global_model.named_parameters() if q.grad is not None}
    got = {}
    ref_cmp = {}
    for name, param in par_model.named_parameters():
        if param.grad is None:
            continue
        g = param.grad
        if isinstance(g, DTensor):
            g = g.full_tensor()
            ref_cmp[name] = ref[name]
        elif g.shape != ref[name].shape:
            # plain local shard (vocab-parallel LM head): compare the slice
            sh = g.shape[0]
            ref_cmp[name] = ref[name].narrow(0, rank * sh, sh)
        else:
            ref_cmp[name] = ref[name]
        got[name] = g
    # TP-sharded grads gathered to full; compare with angle + norm metrics
    assert_grads_close(got, ref_cmp)
    return True


@pytest.mark.distributed
def test_tp2_whole_model_golden():
    assert all(run_distributed(_tp2_whole_model, world_size=2))


def _ep_dp_whole_model(rank, world_size):
    """bench.py --parallelism ep composition: experts sharded over ep, the
    rest replicated over dp, bucketed grad sync; gradients must equal the
    global model run over BOTH ranks' batches (summed)."""
    from torch.distributed.tensor import DTensor

    from d9d_amd.core.dist_context import DeviceMeshParameters
    from d9d_amd.internals.grad_sync import GradientSynchronizer
    from d9d_amd.module.model.qwen3_moe import (
        Qwen3MoEForCausalLM,
        Qwen3MoEModelParameters,
    )
    from d9d_amd.parallel import (
        parallelize_expert_parallel,
        parallelize_replicate,
    )

    p = Qwen3MoEModelParameters.tiny()
    torch.manual_seed(88)
    global_model = Qwen3MoEForCausalLM(p)
    global_model.init_weights()

    torch.manual_seed(88)
    model = Qwen3MoEForCausalLM(p)
    model.init_weights()

    ctx = DeviceMeshParameters(
        data_parallel_replicate=world_size, expert_parallel=world_size
    ).build(device_type="cpu")
    parallelize_expert_parallel(model, ctx.mesh_for("expert"))
    parallelize_replicate(model, ctx.mesh_for("dense"))

    # copy the global weights into the local shards (experts: dim-0 slice)
    gsd = {n: t for n, t in global_model.state_dict().items()}
    with torch.no_grad():
        for name, param in model.named_parameters():
            src = gsd[name]
            local = param.to_local() if isinstance(param, DTensor) else param
            if local.shape == src.shape:
                local.copy_(src)
            else:  # expert shard on dim 0
                e_local = local.shape[0]
                local.copy_(src[rank * e_local : (rank + 1) * e_local])

    sync = GradientSynchronizer(
        list(model.named_parameters()), accumulation_steps=1, bucket_bytes=1 << 20
    )

    torch.manual_seed(600 + rank)
    ids = torch.randint(0, p.vocab_size, (2, 16))
    model(input_ids=ids, labels=ids)["loss"].mean().backward()
    sync.wait()

    # global reference over both ranks' batches
    for r in range(world_size):
        torch.manual_seed(600 + r)
        idr = torch.randint(0, p.vocab_size, (2, 16))
        global_model(input_ids=idr, labels=idr)["loss"].mean().backward()

    for name, param in model.named_parameters():
        if param.grad is None:
            continue
        g = param.grad.to_local() if isinstance(param.grad, DTensor) else param.grad
        ref = gsd[name]  # shapes
        rg = dict(global_model.named_parameters())[name].grad
        if g.shape != rg.shape:  # expert shard
            e_local = g.shape[0]
            rg = rg[rank * e_local : (rank + 1) * e_local]
        ang = grad_angle(g, rg)
        nrm = grad_norm_ratio(g, rg)
        assert ang <= 1e-2, f"{name}: angle {ang:.2e}"
        assert nrm <= 1e-2, f"{name}: norm ratio {nrm:.2e}"
    sync.remove()
    return True


@pytest.mark.distributed
def test_ep_dp_whole_model_golden():
    assert all(run_distributed(_ep_dp_whole_model, world_size=2))
