"""GPU parity tests for the CDNA4 grouped-GEMM kernel."""

import pytest
import torch

from d9d_amd.ops.gmm import gmm, _gmm_loop, _gmm_accum_db


def _rand_sizes(E, total, device="cpu"):
    """Ragged per-expert row counts summing to `total`, incl. zeros."""
    g = torch.Generator().manual_seed(0)
    w = torch.rand(E, generator=g)
    w[::7] = 0  # some empty experts
    sizes = (w / w.sum() * total).long()
    sizes[0] += total - sizes.sum()
    return sizes


@pytest.mark.gpu
@pytest.mark.parametrize(
    "E,total,K,N",
    [
        (8, 4096, 768, 576),
        (128, 32768, 768, 576),
        (16, 1000, 576, 768),   # ragged, non-multiple
        (4, 130, 100, 72),      # tiny + edge tiles
    ],
)
def test_gmm_forward_parity(E, total, K, N):
    device = torch.device("cuda")
    sizes = _rand_sizes(E, total)
    a = torch.randn(total, K, dtype=torch.bfloat16, device=device)
    b = torch.randn(E, K, N, dtype=torch.bfloat16, device=device) * 0.1

    from d9d_amd.ops import _ext

    out = _ext.get_ext().gmm(a, b, sizes)
    ref = _gmm_loop(a.float(), b.float(), sizes)
    torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
@pytest.mark.parametrize("E,total,K,N", [
    (8, 4096, 768, 576),    # BK 256, BN 192
    (8, 4096, 576, 768),    # BK 192, BN 256 (the bench's down-projection wgrad)
    (4, 2048, 1152, 576),   # BK 192, BN 192
    (16, 999, 100, 72),     # legacy fallback (K % 8 != 0)
])
def test_gmm_db_parity(E, total, K, N):
    device = torch.device("cuda")
    sizes = _rand_sizes(E, total)
    a = torch.randn(total, K, dtype=torch.bfloat16, device=device)
    g = torch.randn(total, N, dtype=torch.bfloat16, device=device) * 0.1

    from d9d_amd.ops import _ext

    db = _ext.get_ext().gmm_db(a, g, sizes, E)
    ref = _gmm_accum_db(a.float(), g.float(), sizes, E)
    torch.testing.assert_close(db.float(), ref, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_gmm_autograd_path_uses_kernel():
    device = torch.device("cuda")
    E, total, K, N = 8, 2048, 256, 128
    sizes = _rand_sizes(E, total)
    a = torch.randn(total, K, dtype=torch.bfloat16, device=device, requires_grad=True)
    b = torch.randn(E, K, N, dtype=torch.bfloat16, device=device, requires_grad=True)

    out = gmm(a, b, sizes)
    gr = torch.randn_like(out)
    out.backward(gr)

    a32 = a.detach().float().requires_grad_(True)
    b32 = b.detach().float().requires_grad_(True)
    ref = _gmm_loop(a32, b32, sizes)
    ref.backward(gr.float())

    torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(a.grad.float(), a32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(b.grad.float(), b32.grad, rtol=5e-2, atol=5e-2)


@pytest.mark.gpu
@pytest.mark.parametrize(
    "E,total,K,N",
    [
        (8, 4096, 768, 576),
        (128, 32768, 576, 768),
        (16, 1000, 576, 768),   # ragged, non-multiple
        (4, 130, 100, 72),      # tiny + edge tiles
    ],
)
def test_gmm_nt_forward_parity(E, total, K, N):
    device = torch.device("cuda")
    sizes = _rand_sizes(E, total)
    a = torch.randn(total, K, dtype=torch.bfloat16, device=device)
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device=device) * 0.1
    from d9d_amd.ops import _ext

    out = _ext.get_ext().gmm_nt(a, w, sizes)
    ref = _gmm_loop(a.float(), w.float().transpose(1, 2).contiguous(), sizes)
    torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_gmm_nt_autograd_matches_fp32():
    from d9d_amd.ops import gmm_nt

    device = torch.device("cuda")
    E, total, K, N = 8, 2048, 256, 128
    sizes = _rand_sizes(E, total)
    a = torch.randn(total, K, dtype=torch.bfloat16, device=device, requires_grad=True)
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device=device, requires_grad=True)

    out = gmm_nt(a, w, sizes)
    gr = torch.randn_like(out)
    out.backward(gr)

    a32 = a.detach().float().requires_grad_(True)
    w32 = w.detach().float().requires_grad_(True)
    # reference via autograd on the fp32 loop
    start = 0
    outs = []
    for e, n in enumerate(sizes.tolist()):
        if n:
            outs.append(a32[start : start + n] @ w32[e].t())
        start += n
    ref = torch.cat(outs, dim=0)
    ref.backward(gr.float())

    torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(a.grad.float(), a32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(w.grad.float(), w32.grad, rtol=5e-2, atol=5e-2)
