"""TP and CP exactness tests vs single-process references (gloo ws=2)."""

import pytest
import torch

from tests.helpers import run_distributed


def _make_layer(seed):
    from d9d_amd.module.model.qwen3_dense import (
        Qwen3DenseDecoderLayer,
        Qwen3DenseModelParameters,
    )

    p = Qwen3DenseModelParameters.tiny()  # 4 heads / 2 kv heads
    torch.manual_seed(seed)
    layer = Qwen3DenseDecoderLayer(p)
    layer.reset_parameters()
    return p, layer


def _rotary(p, B, S, offset=0):
    from d9d_amd.module.block.positional import RotaryEmbeddingProvider

    prov = RotaryEmbeddingProvider(rope_dim=p.head_dim, base=p.rope_base)
    pos = torch.arange(offset, offset + S).unsqueeze(0).expand(B, S)
    return prov(pos)


def _tp2_layer(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh

    from d9d_amd.parallel.tensor import parallelize_tensor_parallel

    p, ref_layer = _make_layer(seed=4)
    _, layer = _make_layer(seed=4)  # identical weights

    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("tp",))
    parallelize_tensor_parallel(layer, mesh)

    torch.manual_seed(77)
    x = torch.randn(2, 8, p.hidden_size, requires_grad=True)
    cos_sin = _rotary(p, 2, 8)

    out = layer(x, cos_sin)
    out.sum().backward()

    x_ref = x.detach().clone().requires_grad_(True)
    ref_out = ref_layer(x_ref, cos_sin)
    ref_out.sum().backward()

    torch.testing.assert_close(out, ref_out, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(x.grad, x_ref.grad, rtol=1e-4, atol=1e-5)

    # sharded weight grads match the corresponding slice of the reference grad
    from torch.distributed.tensor import DTensor

    qw = layer.self_attn.q_proj._parameters["weight"]
    assert isinstance(qw, DTensor)
    local_grad = qw.grad.to_local()
    ref_grad = ref_layer.self_attn.q_proj.weight.grad
    expected = ref_grad.chunk(2, dim=0)[rank]
    torch.testing.assert_close(local_grad, expected, rtol=1e-4, atol=1e-5)
    return True


@pytest.mark.distributed
def test_tp2_decoder_layer_exact():
    assert all(run_distributed(_tp2_layer, world_size=2))


def _cp2_attention(rank, world_size):
    import torch.distributed as dist
    from torch.distributed.device_mesh import init_device_mesh

    from d9d_amd.parallel.context import parallelize_context_parallel, shard_sequence

    p, ref_layer = _make_layer(seed=9)
    _, layer = _make_layer(seed=9)

    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("cp_shard",))
    parallelize_context_parallel(layer, mesh)

    torch.manual_seed(55)
    S = 12
    x = torch.randn(2, S, p.hidden_size)
    x_local = shard_sequence(x, rank, 2).requires_grad_(True)

    cos_sin_full = _rotary(p, 2, S)
    cos_sin_local = tuple(shard_sequence(t, rank, 2) for t in cos_sin_full)

    out_local = layer(x_local, cos_sin_local)
    out_local.sum().backward()

    x_ref = x.detach().clone().requires_grad_(True)
    ref_out = ref_layer(x_ref, cos_sin_full)
    ref_out.sum().backward()

    expected_out = shard_sequence(ref_out.detach(), rank, 2)
    torch.testing.assert_close(out_local, expected_out, rtol=1e-4, atol=1e-5)

    expected_xgrad = shard_sequence(x_ref.grad, rank, 2)
    torch.testing.assert_close(x_local.grad, expected_xgrad, rtol=1e-4, atol=1e-5)

    # weight grads: local partials summed over cp == reference grads
    g = layer.self_attn.k_proj.weight.grad.clone()
    dist.all_reduce(g)
    torch.testing.assert_close(
        g, ref_layer.self_attn.k_proj.weight.grad, rtol=1e-4, atol=1e-5
    )
    return True


@pytest.mark.distributed
def test_cp2_attention_exact():
    assert all(run_distributed(_cp2_attention, world_size=2))


def _tp2_sp_layer(rank, world_size):
    """TP + sequence parallel: sequence-sharded activations between blocks;
    outputs and input grads must match the replicated reference."""
    from torch.distributed.device_mesh import init_device_mesh

    from d9d_amd.parallel.tensor import parallelize_tensor_parallel

    p, ref_layer = _make_layer(seed=6)
    _, layer = _make_layer(seed=6)

    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("tp",))
    parallelize_tensor_parallel(layer, mesh, sequence_parallel=True)

    torch.manual_seed(78)
    S = 8
    x = torch.randn(2, S, p.hidden_size)
    cos_sin = _rotary(p, 2, S)

    # local input = this rank's sequence shard
    x_local = x.chunk(2, dim=1)[rank].clone().requires_grad_(True)
    out_local = layer(x_local, cos_sin)
    assert out_local.shape[1] == S // 2
    out_local.sum().backward()

    x_ref = x.detach().clone().requires_grad_(True)
    ref_out = ref_layer(x_ref, cos_sin)
    ref_out.sum().backward()

    torch.testing.assert_close(
        out_local, ref_out.chunk(2, dim=1)[rank], rtol=1e-4, atol=1e-5
    )
    torch.testing.assert_close(
        x_local.grad, x_ref.grad.chunk(2, dim=1)[rank], rtol=1e-4, atol=1e-5
    )
    return True


@pytest.mark.distributed
def test_tp2_sequence_parallel_layer_exact():
    assert all(run_distributed(_tp2_sp_layer, world_size=2))


# -- whole-model TP: embeddings + blocks (incl. MoE) + vocab-parallel head ----


def _tp2_whole_moe_model(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor

    from d9d_amd.module.model.qwen3_moe import (
        Qwen3MoEForCausalLM,
        Qwen3MoEModelParameters,
    )
    from d9d_amd.parallel.tensor import parallelize_tensor_parallel

    p = Qwen3MoEModelParameters.tiny()
    torch.manual_seed(11)
    ref = Qwen3MoEForCausalLM(p)
    ref.init_weights()
    torch.manual_seed(11)
    model = Qwen3MoEForCausalLM(p)
    model.init_weights()

    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("tp",))
    parallelize_tensor_parallel(model, mesh)

    torch.manual_seed(55)
    ids = torch.randint(0, p.vocab_size, (2, 10))
    labels = torch.randint(0, p.vocab_size, (2, 10))
    labels[0, 0] = -100

    out = model(input_ids=ids, labels=labels)
    out["loss"].sum().backward()
    ref_out = ref(input_ids=ids, labels=labels)
    ref_out["loss"].sum().backward()

    torch.testing.assert_close(out["loss"], ref_out["loss"], rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(out["logps"], ref_out["logps"], rtol=1e-4, atol=1e-4)

    # replicated-inside-tp param (router gate): full grad on every rank
    moe = model.model.layers["0"].mlp
    ref_moe = ref.model.layers["0"].mlp
    torch.testing.assert_close(
        moe.router.gate._parameters["weight"].grad,
        ref_moe.router.gate.weight.grad, rtol=1e-4, atol=1e-5,
    )
    # expert shards: gate slice of the packed gate_up grad
    inter = p.intermediate_size
    sh = inter // 2
    gu = moe.experts.gate_up_proj.weight
    ref_gu_grad = ref_moe.experts.gate_up_proj.weight.grad
    expected = torch.cat(
        [ref_gu_grad[:, rank * sh : (rank + 1) * sh],
         ref_gu_grad[:, inter + rank * sh : inter + (rank + 1) * sh]], dim=1)
    torch.testing.assert_close(gu.grad, expected, rtol=1e-4, atol=1e-5)
    # vocab-parallel embedding shard grad
    seg = model.model.embed_tokens.embeddings["regular"]
    ref_seg = ref.model.embed_tokens.embeddings["regular"]
    local = seg._parameters["weight"].grad
    assert isinstance(local, DTensor)
    n = ref_seg.weight.shape[0] // 2
    torch.testing.assert_close(
        local.to_local(), ref_seg.weight.grad[rank * n : (rank + 1) * n],
        rtol=1e-4, atol=1e-5,
    )
    # vocab-parallel head shard grad
    hw = model.lm_head.weights["regular"]
    ref_hw = ref.lm_head.weights["regular"]
    torch.testing.assert_close(
        hw.grad, ref_hw.grad[rank * n : (rank + 1) * n], rtol=1e-4, atol=1e-5
    )
    return True


@pytest.mark.distributed
def test_tp2_whole_moe_model_exact():
    assert all(run_distributed(_tp2_whole_moe_model, world_size=2))


def _tp2_grad_accumulation_exact(rank, world_size):
    """Regression for the round-1 double-reduce bug: replicated-inside-tp
    params (q/k norm, router gate) must accumulate EXACTLY over microbatches."""
    from torch.distributed.device_mesh import init_device_mesh

    from d9d_amd.parallel.tensor import parallelize_tensor_parallel

    p, ref_layer = _make_layer(seed=9)
    _, layer = _make_layer(seed=9)

    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("tp",))
    parallelize_tensor_parallel(layer, mesh)

    cos_sin = _rotary(p, 2, 8)
    torch.manual_seed(31)
    for mb in range(3):  # 3 microbatches, grads accumulate
        x = torch.randn(2, 8, p.hidden_size)
        layer(x, cos_sin).sum().backward()
        ref_layer(x, cos_sin).sum().backward()

    qn = layer.self_attn.q_norm
    ref_qn = ref_layer.self_attn.q_norm
    if qn is not None:
        torch.testing.assert_close(
            qn._parameters["weight"].grad, ref_qn.weight.grad,
            rtol=1e-4, atol=1e-5,
        )
    return True


@pytest.mark.distributed
def test_tp2_grad_accumulation_exact():
    assert all(run_distributed(_tp2_grad_accumulation_exact, world_size=2))


def _tp2_sp_whole_moe_model(rank, world_size):
    """Sequence-parallel whole model: embedding exit reduce-scatters the
    sequence, every block all-gathers/reduce-scatters, head gathers."""
    from torch.distributed.device_mesh import init_device_mesh

    from d9d_amd.module.model.qwen3_moe import (
        Qwen3MoEForCausalLM,
        Qwen3MoEModelParameters,
    )
    from d9d_amd.parallel.tensor import parallelize_tensor_parallel

    p = Qwen3MoEModelParameters.tiny()
    torch.manual_seed(13)
    ref = Qwen3MoEForCausalLM(p)
    ref.init_weights()
    torch.manual_seed(13)
    model = Qwen3MoEForCausalLM(p)
    model.init_weights()

    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("tp",))
    parallelize_tensor_parallel(model, mesh, sequence_parallel=True)

    torch.manual_seed(57)
    ids = torch.randint(0, p.vocab_size, (2, 8))
    labels = torch.randint(0, p.vocab_size, (2, 8))

    out = model(input_ids=ids, labels=labels)
    out["loss"].sum().backward()
    ref_out = ref(input_ids=ids, labels=labels)
    ref_out["loss"].sum().backward()

    torch.testing.assert_close(out["loss"], ref_out["loss"], rtol=1e-4, atol=1e-5)
    moe = model.model.layers["0"].mlp
    ref_moe = ref.model.layers["0"].mlp
    torch.testing.assert_close(
        moe.router.gate._parameters["weight"].grad,
        ref_moe.router.gate.weight.grad, rtol=1e-4, atol=1e-5,
    )
    return True


@pytest.mark.distributed
def test_tp2_sp_whole_moe_model_exact():
    assert all(run_distributed(_tp2_sp_whole_moe_model, world_size=2))


# -- ring attention CP --------------------------------------------------------


def _ring_attention_exact(rank, world_size):
    """Ring attention over contiguous chunks == full causal attention, fwd
    and bwd (GQA shapes)."""
    from d9d_amd.ops.attention import _eager_attention
    from d9d_amd.parallel.context import ring_attention, shard_sequence
    import torch.distributed as dist

    torch.manual_seed(3)
    B, S, Hq, Hkv, D = 2, 16, 4, 2, 16
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)

    # full-sequence reference
    q_ref = q.clone().requires_grad_(True)
    k_ref = k.clone().requires_grad_(True)
    v_ref = v.clone().requires_grad_(True)
    ref, _ = _eager_attention(q_ref, k_ref, v_ref, True, D ** -0.5, (-1, -1), None)
    torch.manual_seed(5)
    dout = torch.randn_like(ref)
    ref.backward(dout)

    # local chunks
    q_l = shard_sequence(q, rank, world_size).requires_grad_(True)
    k_l = shard_sequence(k, rank, world_size).requires_grad_(True)
    v_l = shard_sequence(v, rank, world_size).requires_grad_(True)
    out = ring_attention(q_l, k_l, v_l, group=dist.group.WORLD, causal=True)
    torch.testing.assert_close(
        out, shard_sequence(ref.detach(), rank, world_size), rtol=1e-4, atol=1e-5
    )
    out.backward(shard_sequence(dout, rank, world_size))
    torch.testing.assert_close(
        q_l.grad, shard_sequence(q_ref.grad, rank, world_size), rtol=1e-4, atol=1e-5
    )
    torch.testing.assert_close(
        k_l.grad, shard_sequence(k_ref.grad, rank, world_size), rtol=1e-4, atol=1e-5
    )
    torch.testing.assert_close(
        v_l.grad, shard_sequence(v_ref.grad, rank, world_size), rtol=1e-4, atol=1e-5
    )
    return True


@pytest.mark.distributed
def test_ring_attention_exact_ws2():
    assert all(run_distributed(_ring_attention_exact, world_size=2))


def _cp2_whole_model_golden(rank, world_size):
    """Whole-model ring-CP2: sequence-sharded batch through the parallelized
    model gradient-matches the full-sequence single-process run."""
    from torch.distributed.device_mesh import init_device_mesh

    from d9d_amd.module.model.qwen3_dense import (
        Qwen3DenseForCausalLM,
        Qwen3DenseModelParameters,
    )
    from d9d_amd.parallel.context import (
        parallelize_context_parallel,
        shard_sequence,
    )
    import torch.distributed as dist

    p = Qwen3DenseModelParameters.tiny()
    torch.manual_seed(19)
    ref = Qwen3DenseForCausalLM(p)
    ref.init_weights()
    torch.manual_seed(19)
    model = Qwen3DenseForCausalLM(p)
    model.init_weights()

    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("cp_shard",))
    parallelize_context_parallel(model, mesh)

    torch.manual_seed(23)
    S = 16
    ids = torch.randint(0, p.vocab_size, (2, S))
    labels = torch.randint(0, p.vocab_size, (2, S))

    ref_out = ref(input_ids=ids, labels=labels)
    # per-token sum so sharded losses add exactly across cp ranks
    ref_loss = ref_out["logps"].sum()
    ref_loss.backward()

    out = model(
        input_ids=shard_sequence(ids, rank, 2),
        labels=shard_sequence(labels, rank, 2),
    )
    loss = out["logps"].sum()
    loss.backward()

    # local losses sum to the global loss
    t = loss.detach().clone()
    dist.all_reduce(t)
    torch.testing.assert_close(t, ref_loss.detach(), rtol=1e-4, atol=1e-5)

    # weight grads are partial per cp rank; summed over cp == reference
    for (n, prm), (_, rprm) in zip(
        model.named_parameters(), ref.named_parameters()
    ):
        if prm.grad is None:
            continue
        g = prm.grad.clone()
        dist.all_reduce(g)
        torch.testing.assert_close(
            g, rprm.grad, rtol=2e-4, atol=2e-5,
        )
    return True


@pytest.mark.distributed
def test_cp2_whole_model_golden():
    assert all(run_distributed(_cp2_whole_model_golden, world_size=2))


# -- TP over the other attention families (MLA, GatedDeltaNet) ---------------


def _tp2_mla_exact(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh

    from d9d_amd.module.block.attention import MultiHeadLatentAttention
    from d9d_amd.module.block.positional import RotaryEmbeddingProvider
    from d9d_amd.parallel.tensor import parallelize_tp_mla

    def build():
        torch.manual_seed(21)
        m = MultiHeadLatentAttention(
            64, 4, qk_nope_head_dim=16, qk_rope_head_dim=8, v_head_dim=16,
            kv_lora_rank=32, q_lora_rank=24,
        )
        m.reset_parameters()
        return m

    ref = build()
    mla = build()
    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("tp",))
    parallelize_tp_mla(mla, mesh)

    prov = RotaryEmbeddingProvider(rope_dim=8)
    pos = torch.arange(8).unsqueeze(0).expand(2, 8)
    cos_sin = prov(pos)
    torch.manual_seed(33)
    x = torch.randn(2, 8, 64, requires_grad=True)
    out = mla(x, cos_sin)
    out.sum().backward()

    x_ref = x.detach().clone().requires_grad_(True)
    ref_out = ref(x_ref, cos_sin)
    ref_out.sum().backward()

    torch.testing.assert_close(out, ref_out, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(x.grad, x_ref.grad, rtol=1e-4, atol=1e-5)
    # replicated-inside-region param: full grad everywhere
    torch.testing.assert_close(
        mla.kv_down._parameters["weight"].grad, ref.kv_down.weight.grad,
        rtol=1e-4, atol=1e-5,
    )
    # head-sharded kv_up grad slice
    from torch.distributed.tensor import DTensor

    g = mla.kv_up._parameters["weight"].grad
    assert isinstance(g, DTensor)
    torch.testing.assert_close(
        g.to_local(), ref.kv_up.weight.grad.chunk(2, 0)[rank], rtol=1e-4, atol=1e-5
    )
    return True


@pytest.mark.distributed
def test_tp2_mla_exact():
    assert all(run_distributed(_tp2_mla_exact, world_size=2))


def _tp2_gdn_exact(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh

    from d9d_amd.module.block.attention.linear import GatedDeltaNet
    from d9d_amd.parallel.tensor import parallelize_tp_gdn

    def build():
        torch.manual_seed(23)
        m = GatedDeltaNet(64, num_heads=4, num_kv_heads=2,
                          head_k_dim=16, head_v_dim=16)
        m.reset_parameters()
        return m

    ref = build()
    gdn = build()
    mesh = init_device_mesh("cpu", (2,), mesh_dim_names=("tp",))
    parallelize_tp_gdn(gdn, mesh)

    torch.manual_seed(35)
    x = torch.randn(2, 32, 64, requires_grad=True)
    out = gdn(x)
    out.sum().backward()

    x_ref = x.detach().clone().requires_grad_(True)
    ref_out = ref(x_ref)
    ref_out.sum().backward()

    torch.testing.assert_close(out, ref_out, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(x.grad, x_ref.grad, rtol=1e-4, atol=1e-4)
    # decay-gate bias is head-sharded (the one biased projection)
    from torch.distributed.tensor import DTensor

    b = gdn.decay_gate.proj._parameters["bias"]
    assert isinstance(b, DTensor)
    torch.testing.assert_close(
        b.grad.to_local(), ref.decay_gate.proj.bias.grad.chunk(2, 0)[rank],
        rtol=1e-4, atol=1e-4,
    )
    # shared out_norm weight: full grad everywhere (tp-summed)
    torch.testing.assert_close(
        gdn.out_norm._parameters["weight"].grad, ref.out_norm.weight.grad,
        rtol=1e-4, atol=1e-4,
    )
    return True


@pytest.mark.distributed
def test_tp2_gdn_exact():
    assert all(run_distributed(_tp2_gdn_exact, world_size=2))
