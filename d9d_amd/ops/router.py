"""Fused MoE top-k router op (reference: d9d/module/block/moe/router.py).

GPU: one-wave-per-row HIP kernel doing softmax + biased top-k selection +
renormalization in registers (fwd) and the analytic renorm+softmax backward
(bwd). CPU: the equivalent torch composition.
"""

import torch

from ._ext import get_ext, has_ext


def _torch_router(logits, bias, top_k, renormalize):
    probs = torch.softmax(logits, dim=-1)
    select = probs + bias if bias is not None else probs
    _, indices = torch.topk(select, top_k, dim=-1)
    top_probs = probs.gather(-1, indices)
    if renormalize:
        top_probs = top_probs / top_probs.sum(dim=-1, keepdim=True).clamp_min(1e-20)
    return top_probs, indices


class _RouterTopKFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, bias, top_k, renormalize):
        ext = get_ext()
        top_probs, indices = ext.router_topk_fwd(
            logits.contiguous(), bias, top_k, renormalize
        )
        ctx.save_for_backward(logits, indices)
        ctx.renormalize = renormalize
        return top_probs, indices

    @staticmethod
    def backward(ctx, dtop, _didx):
        logits, indices = ctx.saved_tensors
        dlogits = get_ext().router_topk_bwd(
            logits.contiguous(), indices, dtop.contiguous(), ctx.renormalize
        )
        return dlogits, None, None, None


def router_topk(
    logits: torch.Tensor,
    bias: torch.Tensor | None,
    top_k: int,
    renormalize: bool = True,
) -> tuple[torch.Tensor, torch.Tensor]:
    """logits (T, E) fp32 -> (top_probs (T, k) fp32, indices (T, k) int64).

    `bias` participates in the top-k *selection* only (aux-free load
    balancing) and carries no gradient."""
    if (
        logits.is_cuda
        and has_ext()
        and logits.shape[-1] <= 1024
        and top_k <= 16
    ):
        return _RouterTopKFunction.apply(logits, bias, top_k, renormalize)
    return _torch_router(logits, bias, top_k, renormalize)
