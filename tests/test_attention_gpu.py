"""GPU parity tests for the CDNA4 flash-attention kernel vs the fp32 eager oracle."""

import math

import pytest
import torch

from d9d_amd.ops.attention import _eager_attention, flash_attn_func


@pytest.mark.gpu
def test_mfma_fragment_map_selfcheck():
    from d9d_amd.ops import _ext

    ext = _ext.get_ext()
    device = torch.device("cuda")
    # Asymmetric A and B catch transposed layouts (guide G9).
    a = torch.randn(16, 32, dtype=torch.bfloat16, device=device)
    b = torch.randn(32, 16, dtype=torch.bfloat16, device=device)
    c = ext.mfma_selfcheck(a, b)
    ref = a.float() @ b.float()
    torch.testing.assert_close(c, ref, rtol=2e-2, atol=2e-2)


@pytest.mark.gpu
@pytest.mark.parametrize(
    "B,S,Hq,Hkv,D,causal",
    [
        (2, 256, 8, 2, 128, True),
        (2, 300, 8, 8, 128, True),   # ragged seq, MHA
        (1, 512, 16, 4, 128, False),
        (2, 192, 4, 2, 64, True),    # D=64 template
        (1, 128, 4, 4, 16, True),    # D padded to 32
    ],
)
def test_flash_attn_fwd_parity(B, S, Hq, Hkv, D, causal):
    device = torch.device("cuda")
    q = torch.randn(B, S, Hq, D, dtype=torch.bfloat16, device=device)
    k = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=device)
    v = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=device)
    scale = 1.0 / math.sqrt(D)

    out, lse = flash_attn_func(q, k, v, causal=causal, return_lse=True)
    ref_out, ref_lse = _eager_attention(q, k, v, causal, scale, (-1, -1), None)

    torch.testing.assert_close(out.float(), ref_out.float(), rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(lse, ref_lse, rtol=1e-3, atol=1e-3)


@pytest.mark.gpu
@pytest.mark.parametrize(
    "B,S,Hq,Hkv,D,causal",
    [
        (2, 256, 8, 2, 128, True),
        (1, 300, 4, 4, 128, False),
        (2, 192, 4, 2, 64, True),
    ],
)
def test_flash_attn_bwd_parity(B, S, Hq, Hkv, D, causal):
    device = torch.device("cuda")
    q = torch.randn(B, S, Hq, D, dtype=torch.bfloat16, device=device, requires_grad=True)
    k = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=device, requires_grad=True)
    v = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=device, requires_grad=True)
    dout = torch.randn(B, S, Hq, D, dtype=torch.bfloat16, device=device)

    out = flash_attn_func(q, k, v, causal=causal)
    out.backward(dout)

    q32 = q.detach().float().requires_grad_(True)
    k32 = k.detach().float().requires_grad_(True)
    v32 = v.detach().float().requires_grad_(True)
    ref_out, _ = _eager_attention(
        q32, k32, v32, causal, 1.0 / math.sqrt(D), (-1, -1), None
    )
    ref_out.backward(dout.float())

    torch.testing.assert_close(q.grad.float(), q32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(k.grad.float(), k32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(v.grad.float(), v32.grad, rtol=5e-2, atol=5e-2)


@pytest.mark.gpu
def test_flash_attn_long_seq_smoke():
    device = torch.device("cuda")
    B, S, Hq, Hkv, D = 2, 4096, 16, 4, 128
    q = torch.randn(B, S, Hq, D, dtype=torch.bfloat16, device=device, requires_grad=True)
    k = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=device, requires_grad=True)
    v = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=device, requires_grad=True)
    out = flash_attn_func(q, k, v, causal=True)
    out.sum().backward()
    torch.cuda.synchronize()
    assert torch.isfinite(out.float()).all()
    assert torch.isfinite(q.grad.float()).all()


@pytest.mark.gpu
def test_flash_attn_q_offset_cp_parity():
    """CP decomposition: local q block + full KV + q_offset == full attention."""
    device = torch.device("cuda")
    B, S, Hq, Hkv, D = 2, 256, 8, 2, 128
    q = torch.randn(B, S, Hq, D, dtype=torch.bfloat16, device=device)
    k = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=device)
    v = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=device)
    full = flash_attn_func(q, k, v, causal=True)
    half = S // 2
    for r in range(2):
        part = flash_attn_func(
            q[:, r * half : (r + 1) * half], k, v, causal=True, q_offset=r * half
        )
        torch.testing.assert_close(
            part.float(), full[:, r * half : (r + 1) * half].float(),
            rtol=2e-2, atol=2e-2,
        )


@pytest.mark.gpu
def test_flash_sliding_window_matches_eager():
    torch.manual_seed(11)
    B, S, Hq, Hkv, D = 2, 512, 8, 2, 64
    q = torch.randn(B, S, Hq, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    k = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    v = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)

    out = flash_attn_func(q, k, v, causal=True, window_size=(128, -1))
    g = torch.randn_like(out)
    out.backward(g)

    q32 = q.detach().float().requires_grad_(True)
    k32 = k.detach().float().requires_grad_(True)
    v32 = v.detach().float().requires_grad_(True)
    ref, _ = _eager_attention(q32, k32, v32, True, q.shape[-1] ** -0.5, (128, -1), None)
    ref.backward(g.float())

    torch.testing.assert_close(out.float(), ref.float(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(q.grad.float(), q32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(k.grad.float(), k32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(v.grad.float(), v32.grad, rtol=5e-2, atol=5e-2)


@pytest.mark.gpu
def test_flash_attention_sinks_matches_eager():
    torch.manual_seed(12)
    B, S, Hq, Hkv, D = 2, 256, 8, 2, 64
    q = torch.randn(B, S, Hq, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    k = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    v = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    sinks = torch.randn(Hq, dtype=torch.float32, device="cuda", requires_grad=True)

    out, lse = flash_attn_func(q, k, v, causal=True, sinks=sinks, return_lse=True)
    g = torch.randn_like(out)
    out.backward(g)

    q32 = q.detach().float().requires_grad_(True)
    k32 = k.detach().float().requires_grad_(True)
    v32 = v.detach().float().requires_grad_(True)
    s32 = sinks.detach().clone().requires_grad_(True)
    ref, ref_lse = _eager_attention(q32, k32, v32, True, q.shape[-1] ** -0.5, (-1, -1), s32)
    ref.backward(g.float())

    torch.testing.assert_close(out.float(), ref.float(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(lse, ref_lse, rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(q.grad.float(), q32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(k.grad.float(), k32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(v.grad.float(), v32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(sinks.grad, s32.grad, rtol=5e-2, atol=5e-2)


@pytest.mark.gpu
def test_flash_varlen_matches_eager():
    from d9d_amd.ops.attention import _eager_varlen, flash_attn_varlen_func

    torch.manual_seed(13)
    lens_q = [100, 260, 37, 512]
    lens_k = [100, 260, 37, 512]
    cu_q = torch.tensor([0] + list(torch.tensor(lens_q).cumsum(0)), dtype=torch.int32)
    cu_k = torch.tensor([0] + list(torch.tensor(lens_k).cumsum(0)), dtype=torch.int32)
    Hq, Hkv, D = 8, 2, 64
    tq, tk = int(cu_q[-1]), int(cu_k[-1])
    q = torch.randn(tq, Hq, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    k = torch.randn(tk, Hkv, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    v = torch.randn(tk, Hkv, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)

    out, lse = flash_attn_varlen_func(
        q, k, v, cu_q.cuda(), cu_k.cuda(), causal=True, return_lse=True
    )
    g = torch.randn_like(out)
    out.backward(g)

    q32 = q.detach().float().requires_grad_(True)
    k32 = k.detach().float().requires_grad_(True)
    v32 = v.detach().float().requires_grad_(True)
    ref, ref_lse = _eager_varlen(q32, k32, v32, cu_q, cu_k, True, D ** -0.5, (-1, -1), None)
    ref.backward(g.float())

    torch.testing.assert_close(out.float(), ref.float(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(lse, ref_lse, rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(q.grad.float(), q32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(k.grad.float(), k32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(v.grad.float(), v32.grad, rtol=5e-2, atol=5e-2)


@pytest.mark.gpu
def test_flash_varlen_cross_attention_lengths():
    """len_q != len_k: causal aligns sequence ends."""
    from d9d_amd.ops.attention import _eager_varlen, flash_attn_varlen_func

    torch.manual_seed(14)
    lens_q = [64, 130]
    lens_k = [200, 300]
    cu_q = torch.tensor([0] + list(torch.tensor(lens_q).cumsum(0)), dtype=torch.int32)
    cu_k = torch.tensor([0] + list(torch.tensor(lens_k).cumsum(0)), dtype=torch.int32)
    Hq, Hkv, D = 4, 4, 128
    q = torch.randn(int(cu_q[-1]), Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(int(cu_k[-1]), Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(int(cu_k[-1]), Hkv, D, dtype=torch.bfloat16, device="cuda")

    out = flash_attn_varlen_func(q, k, v, cu_q.cuda(), cu_k.cuda(), causal=True)
    ref, _ = _eager_varlen(
        q.float(), k.float(), v.float(), cu_q, cu_k, True, D ** -0.5, (-1, -1), None
    )
    torch.testing.assert_close(out.float(), ref.float(), rtol=3e-2, atol=3e-2)
