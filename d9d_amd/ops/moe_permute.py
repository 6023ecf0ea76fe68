"""MoE token permutation (reference: d9d/kernel/moe/permute_with_probs.py,
indices_to_multihot.py).

`moe_permute`: replicate each token once per selected expert and sort the
replicas by expert id, so each expert's rows are contiguous for the grouped
GEMM. `moe_unpermute`: sum a token's replicas back (prob-weighted).

GPU: dedicated CDNA4 gather/combine kernels (csrc/moe_permute.hip). The
combine is a deterministic CSR gather over each token's K replica rows via
the inverse permutation (flat replica r of token t is t*K + j) — no atomics.
CPU: index-op reference path.
"""

import torch

from ._ext import get_ext, has_ext


def _use_kernels(t: torch.Tensor) -> bool:
    return t.is_cuda and has_ext()


class _GatherRowsFunction(torch.autograd.Function):
    """permuted[r] = tokens[row_to_token[r]]; backward combines replicas."""

    @staticmethod
    def forward(ctx, tokens, row_to_token, inv, num_tokens, top_k):
        ctx.save_for_backward(row_to_token, inv)
        ctx.num_tokens = num_tokens
        ctx.top_k = top_k
        if _use_kernels(tokens):
            return get_ext().moe_gather_rows(tokens.contiguous(), row_to_token, None)
        return tokens.index_select(0, row_to_token)

    @staticmethod
    def backward(ctx, grad):
        row_to_token, inv = ctx.saved_tensors
        if _use_kernels(grad):
            d_tokens = get_ext().moe_csr_combine(
                grad.contiguous(), None, inv, ctx.num_tokens, ctx.top_k
            )
        else:
            d_tokens = torch.zeros(
                (ctx.num_tokens, grad.shape[1]), dtype=grad.dtype, device=grad.device
            ).index_add(0, row_to_token, grad)
        return d_tokens, None, None, None, None


class _WeightedCombineFunction(torch.autograd.Function):
    """out[t] = sum_j probs[r_j] * expert_out[r_j] over token t's replicas."""

    @staticmethod
    def forward(ctx, expert_out, probs, row_to_token, inv, num_tokens, top_k):
        ctx.save_for_backward(expert_out, probs, row_to_token, inv)
        if _use_kernels(expert_out):
            return get_ext().moe_csr_combine(
                expert_out.contiguous(), probs.float().contiguous(), inv,
                num_tokens, top_k,
            )
        weighted = expert_out * probs.unsqueeze(-1).to(expert_out.dtype)
        out = torch.zeros(
            (num_tokens, expert_out.shape[1]),
            dtype=expert_out.dtype, device=expert_out.device,
        )
        return out.index_add(0, row_to_token, weighted)

    @staticmethod
    def backward(ctx, grad):
        expert_out, probs, row_to_token, inv = ctx.saved_tensors
        grad = grad.contiguous()
        if _use_kernels(grad):
            d_expert = get_ext().moe_gather_rows(
                grad, row_to_token, probs.float().contiguous()
            )
            d_probs = get_ext().moe_row_dot(grad, expert_out.contiguous(), row_to_token)
        else:
            g_rows = grad.index_select(0, row_to_token)
            d_expert = g_rows * probs.unsqueeze(-1).to(grad.dtype)
            d_probs = (g_rows.float() * expert_out.float()).sum(-1)
        return d_expert, d_probs.to(probs.dtype), None, None, None, None


def moe_permute(
    tokens: torch.Tensor,  # (T, H)
    expert_indices: torch.Tensor,  # (T, K) int64
    probs: torch.Tensor,  # (T, K)
    num_experts: int,
):
    """Returns (permuted_tokens (T*K, H), permuted_probs (T*K,),
    permute_ctx, tokens_per_expert (E,))."""
    T, K = expert_indices.shape
    flat_experts = expert_indices.reshape(-1)  # (T*K,)
    order = torch.argsort(flat_experts, stable=True)
    inv = torch.empty_like(order)
    inv[order] = torch.arange(order.numel(), device=order.device)
    row_to_token = order // K  # source token per permuted row
    permuted_tokens = _GatherRowsFunction.apply(tokens, row_to_token, inv, T, K)
    permuted_probs = probs.reshape(-1).index_select(0, order)
    tokens_per_expert = torch.bincount(flat_experts, minlength=num_experts)
    return permuted_tokens, permuted_probs, (row_to_token, inv, T, K), tokens_per_expert


def moe_unpermute(
    expert_out: torch.Tensor,  # (T*K, H)
    permuted_probs: torch.Tensor,  # (T*K,)
    permute_ctx,
) -> torch.Tensor:
    """Weighted sum of each token's expert outputs back to (T, H)."""
    row_to_token, inv, T, K = permute_ctx
    return _WeightedCombineFunction.apply(
        expert_out, permuted_probs, row_to_token, inv, T, K
    )
