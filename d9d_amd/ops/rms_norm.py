"""RMSNorm op: CDNA4 HIP kernel on GPU, fp32 torch reference on CPU.

Mirrors the reference op contract (d9d/kernel/normalization/rms):
fp32 math, optional zero-centered weight (w+1), inv_rms saved for backward.
"""

import torch

from ._ext import get_ext


def _rms_norm_ref_fwd(x: torch.Tensor, w: torch.Tensor, eps: float, zero_centered: bool):
    x32 = x.float()
    inv_rms = torch.rsqrt(x32.pow(2).mean(dim=-1) + eps)
    w32 = w.float() + (1.0 if zero_centered else 0.0)
    y = x32 * inv_rms.unsqueeze(-1) * w32
    return y.to(x.dtype), inv_rms


def _rms_norm_ref_bwd(x, w, dy, inv_rms, zero_centered):
    x32 = x.float()
    dy32 = dy.float()
    w32 = w.float() + (1.0 if zero_centered else 0.0)
    x_hat = x32 * inv_rms.unsqueeze(-1)
    t = dy32 * w32
    s = (t * x_hat).mean(dim=-1, keepdim=True)
    dx = inv_rms.unsqueeze(-1) * (t - x_hat * s)
    dw = (dy32 * x_hat).reshape(-1, x.shape[-1]).sum(dim=0)
    return dx.to(x.dtype), dw


class _RMSNormFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps, zero_centered):
        if x.is_cuda:
            ext = get_ext()
            x2d = x.contiguous()
            y, inv_rms = ext.rms_norm_fwd(x2d, w.contiguous(), eps, zero_centered)
        else:
            y, inv_rms = _rms_norm_ref_fwd(x, w, eps, zero_centered)
        ctx.save_for_backward(x, w, inv_rms)
        ctx.zero_centered = zero_centered
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, inv_rms = ctx.saved_tensors
        if x.is_cuda:
            ext = get_ext()
            dx, dw = ext.rms_norm_bwd(
                x.contiguous(), w.contiguous(), dy.contiguous(), inv_rms,
                ctx.zero_centered,
            )
        else:
            dx, dw = _rms_norm_ref_bwd(x, w, dy, inv_rms, ctx.zero_centered)
        return dx, dw.to(w.dtype), None, None


def rms_norm(
    x: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    zero_centered: bool = False,
) -> torch.Tensor:
    return _RMSNormFunction.apply(x, weight, eps, zero_centered)
