"""Fused silu(a)*b op (SwiGLU activation; reference: d9d/kernel/swiglu)."""

import torch
import torch.nn.functional as F

from ._ext import get_ext


class _SiluMulFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        ctx.save_for_backward(a, b)
        if a.is_cuda:
            return get_ext().silu_mul_fwd(a.contiguous(), b.contiguous())
        a32 = a.float()
        return (F.silu(a32) * b.float()).to(a.dtype)

    @staticmethod
    def backward(ctx, g):
        a, b = ctx.saved_tensors
        if a.is_cuda:
            da, db = get_ext().silu_mul_bwd(a.contiguous(), b.contiguous(), g.contiguous())
            return da, db
        a32, b32, g32 = a.float(), b.float(), g.float()
        sig = torch.sigmoid(a32)
        da = g32 * b32 * sig * (1 + a32 * (1 - sig))
        db = g32 * a32 * sig
        return da.to(a.dtype), db.to(b.dtype)


def silu_mul(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return _SiluMulFunction.apply(a, b)


class _SiluMulPackedFunction(torch.autograd.Function):
    """silu(x[:, :I]) * x[:, I:] for x = [gate | up] from a fused projection."""

    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        if x.is_cuda:
            return get_ext().silu_mul_packed_fwd(x.contiguous())
        inter = x.shape[-1] // 2
        a32, b32 = x[:, :inter].float(), x[:, inter:].float()
        return (F.silu(a32) * b32).to(x.dtype)

    @staticmethod
    def backward(ctx, g):
        (x,) = ctx.saved_tensors
        if x.is_cuda:
            return get_ext().silu_mul_packed_bwd(x.contiguous(), g.contiguous())
        inter = x.shape[-1] // 2
        a32, b32, g32 = x[:, :inter].float(), x[:, inter:].float(), g.float()
        sig = torch.sigmoid(a32)
        da = g32 * b32 * sig * (1 + a32 * (1 - sig))
        db = g32 * a32 * sig
        return torch.cat([da, db], dim=-1).to(x.dtype)


def silu_mul_packed(x: torch.Tensor) -> torch.Tensor:
    return _SiluMulPackedFunction.apply(x)
