"""Tests for MLA, GatedDeltaNet, SDPA family, hidden-state aggregator."""

import torch

from d9d_amd.module.block.attention import (
    GatedDeltaNet,
    MultiHeadLatentAttention,
    build_sdpa_backend,
    chunk_gated_delta_rule,
)
from d9d_amd.module.block.attention.sdpa import EagerSdpaConfig
from d9d_amd.module.block.hidden_states_aggregator import HiddenStatesAggregator
from d9d_amd.module.block.positional import RotaryEmbeddingProvider


def test_mla_forward_backward_shapes():
    mla = MultiHeadLatentAttention(
        64, 4, qk_nope_head_dim=16, qk_rope_head_dim=8, v_head_dim=16,
        kv_lora_rank=32, q_lora_rank=24,
    )
    mla.reset_parameters()
    prov = RotaryEmbeddingProvider(rope_dim=8)
    pos = torch.arange(10).unsqueeze(0).expand(2, 10)
    x = torch.randn(2, 10, 64, requires_grad=True)
    out = mla(x, prov(pos))
    assert out.shape == (2, 10, 64)
    out.sum().backward()
    assert x.grad is not None
    assert mla.kv_up.weight.grad is not None


def test_delta_rule_is_causal_and_decays():
    torch.manual_seed(0)
    B, H, S, D = 1, 2, 8, 4
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    beta = torch.sigmoid(torch.randn(B, H, S))
    g = torch.zeros(B, H, S)  # no decay

    out = chunk_gated_delta_rule(q, k, v, beta, g)
    # causality: changing future inputs must not change earlier outputs
    v2 = v.clone()
    v2[:, :, -1] += 100
    out2 = chunk_gated_delta_rule(q, k, v2, beta, g)
    torch.testing.assert_close(out[:, :, :-1], out2[:, :, :-1])
    assert not torch.allclose(out[:, :, -1], out2[:, :, -1])

    # full decay wipes the state: each output only sees its own write
    g_full = torch.full((B, H, S), -50.0)
    out3 = chunk_gated_delta_rule(q, k, v, beta, g_full)
    t = 3
    expected = torch.einsum(
        "bhk,bhkv->bhv",
        q[:, :, t],
        torch.einsum("bhk,bhv->bhkv", k[:, :, t], beta[:, :, t, None] * v[:, :, t]),
    )
    torch.testing.assert_close(out3[:, :, t], expected.to(out3.dtype), rtol=1e-4, atol=1e-5)


def test_gdn_module_runs():
    gdn = GatedDeltaNet(32, num_heads=4, num_kv_heads=2, head_k_dim=8, head_v_dim=8)
    gdn.reset_parameters()
    x = torch.randn(2, 12, 32, requires_grad=True)
    out = gdn(x)
    out.sum().backward()
    assert out.shape == x.shape
    assert x.grad is not None


def test_sdpa_factory_env_and_config(monkeypatch):
    import math

    backend = build_sdpa_backend(EagerSdpaConfig())
    q = torch.randn(1, 8, 2, 16)
    k = torch.randn(1, 8, 2, 16)
    v = torch.randn(1, 8, 2, 16)
    out = backend(q, k, v, causal=True)
    from d9d_amd.ops.attention import _eager_attention

    ref, _ = _eager_attention(q, k, v, True, 1 / math.sqrt(16), (-1, -1), None)
    torch.testing.assert_close(out, ref)

    monkeypatch.setenv("D9D_BACKEND_AUTO_SDPA", "eager")
    assert build_sdpa_backend(None).__name__ == "_eager"
    monkeypatch.delenv("D9D_BACKEND_AUTO_SDPA")
    assert build_sdpa_backend(None).__name__ == "_cdna4_flash"


def test_hidden_states_aggregator():
    agg = HiddenStatesAggregator("mean")
    h = torch.randn(2, 6, 8)
    a = agg.initial(h)
    a = agg.append(a, h)
    a = agg.append(a, h * 2)
    assert a.shape == (2, 2, 8)
    torch.testing.assert_close(a[0], h.mean(1))
    assert HiddenStatesAggregator("none").append(None, h) is None


def test_chunked_delta_rule_matches_step_oracle():
    import torch.nn.functional as F

    from d9d_amd.module.block.attention.linear.gated_deltanet import (
        step_gated_delta_rule,
    )

    torch.manual_seed(1)
    B, H, S, Dk, Dv = 2, 3, 130, 16, 24
    q = F.normalize(torch.randn(B, H, S, Dk), dim=-1)
    k = F.normalize(torch.randn(B, H, S, Dk), dim=-1)
    v = torch.randn(B, H, S, Dv)
    beta = torch.rand(B, H, S)
    g = -torch.rand(B, H, S) * 0.2

    ref = step_gated_delta_rule(q, k, v, beta, g)
    for cs in (32, 64, 130):
        got = chunk_gated_delta_rule(q, k, v, beta, g, chunk_size=cs)
        torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-5)


def test_chunked_delta_rule_backward_flows():
    import torch.nn.functional as F

    torch.manual_seed(2)
    B, H, S, Dk, Dv = 1, 2, 64, 8, 8
    q = F.normalize(torch.randn(B, H, S, Dk), dim=-1).requires_grad_(True)
    k = F.normalize(torch.randn(B, H, S, Dk), dim=-1).requires_grad_(True)
    v = torch.randn(B, H, S, Dv, requires_grad=True)
    beta = torch.rand(B, H, S, requires_grad=True)
    g = (-torch.rand(B, H, S) * 0.2).requires_grad_(True)
    out = chunk_gated_delta_rule(q, k, v, beta, g, chunk_size=16)
    out.sum().backward()
    for t in (q, k, v, beta, g):
        assert t.grad is not None and torch.isfinite(t.grad).all()


import pytest


@pytest.mark.gpu
def test_gated_deltanet_module_gpu():
    torch.manual_seed(5)
    m = GatedDeltaNet(64, num_heads=4, head_k_dim=16, head_v_dim=16).cuda().bfloat16()
    m.reset_parameters()
    x = torch.randn(2, 128, 64, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    out = m(x)
    assert out.shape == (2, 128, 64)
    out.float().sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad.float()).all()


@pytest.mark.gpu
def test_mla_module_gpu():
    torch.manual_seed(6)
    mla = MultiHeadLatentAttention(
        64, 4, qk_nope_head_dim=16, qk_rope_head_dim=8, v_head_dim=16,
        kv_lora_rank=32, q_lora_rank=24,
    ).cuda().bfloat16()
    mla.reset_parameters()
    prov = RotaryEmbeddingProvider(rope_dim=8).cuda()
    pos = torch.arange(32).unsqueeze(0).expand(2, 32).cuda()
    x = torch.randn(2, 32, 64, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    cos_sin = prov(pos)
    out = mla(x, cos_sin)
    out.float().sum().backward()
    assert x.grad is not None


def test_gqa_packed_documents_matches_per_doc():
    """cu_seqlens path == running each document separately."""
    from d9d_amd.module.block.attention import GroupedQueryAttention

    torch.manual_seed(9)
    attn = GroupedQueryAttention(32, 4, 2, 8, use_qk_norm=False)
    attn.reset_parameters()
    prov = RotaryEmbeddingProvider(rope_dim=8)
    lens = [5, 9, 3]
    cu = torch.tensor([0, 5, 14, 17], dtype=torch.int32)
    xs = [torch.randn(1, n, 32) for n in lens]
    packed = torch.cat(xs, dim=1)

    # rotary positions restart per document
    pos = torch.cat([torch.arange(n) for n in lens]).unsqueeze(0)
    out_packed = attn(packed, prov(pos), cu_seqlens=cu)

    outs = []
    for x, n in zip(xs, lens):
        p = torch.arange(n).unsqueeze(0)
        outs.append(attn(x, prov(p)))
    ref = torch.cat(outs, dim=1)
    torch.testing.assert_close(out_packed, ref, rtol=1e-4, atol=1e-5)


@pytest.mark.gpu
def test_gqa_packed_documents_gpu():
    from d9d_amd.module.block.attention import GroupedQueryAttention

    torch.manual_seed(10)
    attn = GroupedQueryAttention(64, 4, 2, 16, use_qk_norm=False).cuda().bfloat16()
    attn.reset_parameters()
    prov = RotaryEmbeddingProvider(rope_dim=16).cuda()
    lens = [100, 260, 37]
    cu = torch.tensor([0, 100, 360, 397], dtype=torch.int32, device="cuda")
    pos = torch.cat([torch.arange(n) for n in lens]).unsqueeze(0).cuda()
    x = torch.randn(1, 397, 64, device="cuda", dtype=torch.bfloat16, requires_grad=True)

    out = attn(x, prov(pos), cu_seqlens=cu)
    out.float().sum().backward()
    assert x.grad is not None

    # parity: each document processed separately
    with torch.no_grad():
        outs = []
        for s, e in zip(cu[:-1].tolist(), cu[1:].tolist()):
            p = torch.arange(e - s).unsqueeze(0).cuda()
            outs.append(attn(x[:, s:e].detach(), prov(p)))
        ref = torch.cat(outs, dim=1)
    torch.testing.assert_close(out.float(), ref.float(), rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_gdn_chunk_kernel_vs_oracle_gpu():
    """Fused CDNA4 chunk kernel vs the per-timestep fp32 oracle (bf16
    GEMM-operand tolerance) and vs the torch WY path."""
    import torch.nn.functional as F

    from d9d_amd.module.block.attention.linear.gated_deltanet import (
        _chunk_gated_delta_rule_torch,
        step_gated_delta_rule,
    )
    from d9d_amd.ops._ext import get_ext

    torch.manual_seed(7)
    B, H, S, D = 2, 3, 200, 64
    q = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).bfloat16()
    k = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).bfloat16()
    v = (torch.randn(B, H, S, D, device="cuda") * 0.5).bfloat16()
    beta = torch.rand(B, H, S, device="cuda")
    g = -torch.rand(B, H, S, device="cuda") * 0.2

    out = get_ext().gdn_chunk_fwd(q, k, v, beta, g, False, False, False)[0]
    oracle = step_gated_delta_rule(q.float(), k.float(), v.float(), beta, g)
    wy = _chunk_gated_delta_rule_torch(q.float(), k.float(), v.float(), beta, g)
    torch.testing.assert_close(wy, oracle.to(wy.dtype), rtol=1e-3, atol=1e-3)
    # kernel runs the chunk GEMMs in bf16 (state cast per chunk, fla-style)
    torch.testing.assert_close(out.float(), oracle.float(), rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_gdn_chunk_kernel_final_state_gpu():
    import torch.nn.functional as F

    from d9d_amd.ops._ext import get_ext

    torch.manual_seed(8)
    B, H, S, D = 1, 2, 64, 64
    q = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).bfloat16()
    k = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).bfloat16()
    v = (torch.randn(B, H, S, D, device="cuda") * 0.5).bfloat16()
    beta = torch.rand(B, H, S, device="cuda")
    g = -torch.rand(B, H, S, device="cuda") * 0.1

    out, fs = get_ext().gdn_chunk_fwd(q, k, v, beta, g, True, False, False)
    assert fs.shape == (B, H, D, D)
    assert torch.isfinite(fs).all()

    # fp32 reference state
    state = torch.zeros(B, H, D, D, device="cuda")
    q32, k32, v32 = q.float(), k.float(), v.float()
    for t in range(S):
        kt, vt = k32[:, :, t], v32[:, :, t]
        bt = beta[:, :, t].unsqueeze(-1)
        gt = g[:, :, t].exp().unsqueeze(-1).unsqueeze(-1)
        state = state * gt
        pred = torch.einsum("bhk,bhkv->bhv", kt, state)
        state = state + torch.einsum("bhk,bhv->bhkv", kt, bt * (vt - pred))
    torch.testing.assert_close(fs, state, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_causal_conv_silu_kernel_gpu():
    """Fused conv+SiLU kernel vs the eager pad/conv1d/silu chain, fwd+bwd."""
    import torch.nn.functional as F

    from d9d_amd.ops._ext import get_ext

    torch.manual_seed(9)
    B, S, C, K = 2, 37, 48, 4
    x = torch.randn(B, S, C, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(C, K, device="cuda", dtype=torch.bfloat16, requires_grad=True)

    out = get_ext().causal_conv_silu_fwd(x, w)

    x32 = x.detach().float().requires_grad_(True)
    w32 = w.detach().float().requires_grad_(True)
    xt = F.pad(x32.transpose(1, 2), (K - 1, 0))
    ref = F.silu(F.conv1d(xt, w32.unsqueeze(1), groups=C).transpose(1, 2))
    torch.testing.assert_close(out.float(), ref, rtol=2e-2, atol=2e-2)

    dy = torch.randn_like(out)
    dx, dw = get_ext().causal_conv_silu_bwd(x, w, dy)
    ref.backward(dy.float())
    torch.testing.assert_close(dx.float(), x32.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(dw.float(), w32.grad, rtol=3e-2, atol=2e-1)


@pytest.mark.gpu
def test_gdn_module_uses_kernel_and_trains_gpu():
    """Whole GatedDeltaNet module on the kernel path (Dk=Dv=64):
    forward+backward finite, and grads close to the torch-path module."""
    torch.manual_seed(11)
    m = GatedDeltaNet(128, num_heads=2, head_k_dim=64, head_v_dim=64).cuda().bfloat16()
    m.reset_parameters()
    x = torch.randn(2, 96, 128, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    out = m(x)
    out.float().sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad.float()).all()
    for n, p in m.named_parameters():
        assert p.grad is None or torch.isfinite(p.grad.float()).all(), n


def test_gdn_derived_backward_matches_autograd():
    """The hand-derived chunk backward (_chunk_gdn_backward) matches
    autograd through the torch WY graph, including a padded tail chunk."""
    import torch

    from d9d_amd.module.block.attention.linear.gated_deltanet import (
        _chunk_gated_delta_rule_torch,
        _chunk_gdn_backward,
    )

    torch.manual_seed(3)
    for B, H, S, Dk, Dv, C in [(2, 2, 96, 32, 48, 64), (1, 3, 64, 16, 16, 32)]:
        q = torch.randn(B, H, S, Dk, requires_grad=True)
        k = torch.nn.functional.normalize(
            torch.randn(B, H, S, Dk), dim=-1
        ).requires_grad_(True)
        v = torch.randn(B, H, S, Dv, requires_grad=True)
        b = torch.rand(B, H, S, requires_grad=True)
        g = (-torch.rand(B, H, S) * 0.2).requires_grad_(True)
        out = _chunk_gated_delta_rule_torch(q, k, v, b, g, C)
        dout = torch.randn_like(out)
        ref = torch.autograd.grad(out, (q, k, v, b, g), dout)
        mine = _chunk_gdn_backward(
            q.detach(), k.detach(), v.detach(), b.detach(), g.detach(), dout, C
        )
        for name, a, m in zip("qkvbg", ref, mine):
            torch.testing.assert_close(
                a, m, rtol=2e-4, atol=2e-4,
                msg=lambda s, n=name: f"d{n}: {s}",
            )


def test_torch_sdpa_backend_matches_eager():
    """The torch-SDPA backend (reference TorchSdpa analog) matches the
    eager reference, including GQA and sliding window."""
    import torch

    from d9d_amd.module.block.attention.sdpa import (
        TorchSdpaConfig,
        build_sdpa_backend,
        _eager,
    )

    torch.manual_seed(5)
    be = build_sdpa_backend(TorchSdpaConfig())
    B, S, Hq, Hkv, D = 2, 33, 4, 2, 16
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    for kw in [dict(causal=True), dict(causal=False),
               dict(causal=True, window_size=(7, 0))]:
        out = be(q, k, v, **kw)
        ref = _eager(q, k, v, **kw)
        torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)


@pytest.mark.gpu
@pytest.mark.parametrize("B,H", [(2, 3), (16, 13)])  # DVT=32 split / DVT=64
def test_gdn_scan_kernels_match_torch_scans_gpu(B, H):
    """The CDNA4 backward scan kernels (both Dv-split instantiations)
    produce the same gradients as the torch scan loops."""
    import os

    import torch
    import torch.nn.functional as F

    from d9d_amd.module.block.attention.linear.gated_deltanet import (
        _chunk_gdn_backward,
    )

    torch.manual_seed(9)
    S, D = 256, 64
    q = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).float()
    k = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).float()
    v = (torch.randn(B, H, S, D, device="cuda") * 0.5).float()
    beta = torch.rand(B, H, S, device="cuda")
    g = -torch.rand(B, H, S, device="cuda") * 0.2
    do = torch.randn(B, H, S, D, device="cuda")

    os.environ["D9D_GDN_BWD_SCAN"] = "0"
    try:
        ref = _chunk_gdn_backward(q, k, v, beta, g, do)
    finally:
        os.environ["D9D_GDN_BWD_SCAN"] = "1"
    out = _chunk_gdn_backward(q, k, v, beta, g, do)
    for name, a, b in zip("qkvbg", ref, out):
        torch.testing.assert_close(
            a.float(), b.float(), rtol=2e-2, atol=2e-2,
            msg=lambda s, n=name: f"d{n}: {s}",
        )
