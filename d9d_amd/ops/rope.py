"""Fused rotary embedding op (HALF layout): rotates q and k in one pass.

GPU: csrc/rope.hip; CPU falls back to the eager composition in
module/block/positional/rope.py. Backward = inverse rotation (sin -> -sin).
"""

import torch

from ._ext import get_ext, has_ext


class _RopeQKFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, cos, sin):
        ctx.save_for_backward(cos, sin)
        q_out, k_out = get_ext().rope_qk(
            q.contiguous(), k.contiguous(), cos.contiguous(), sin.contiguous(), 1.0
        )
        return q_out, k_out

    @staticmethod
    def backward(ctx, dq, dk):
        cos, sin = ctx.saved_tensors
        dq_in, dk_in = get_ext().rope_qk(
            dq.contiguous(), dk.contiguous(), cos, sin, -1.0
        )
        return dq_in, dk_in, None, None


def rope_qk_available(q: torch.Tensor, rope_dim: int) -> bool:
    return q.is_cuda and q.dtype == torch.bfloat16 and rope_dim % 4 == 0 and has_ext()


def rope_qk(q, k, cos, sin):
    """q (B,S,Hq,D), k (B,S,Hkv,D), cos/sin (B,S,rope_dim) fp32 (halves duplicated)."""
    return _RopeQKFunction.apply(q, k, cos.float(), sin.float())
