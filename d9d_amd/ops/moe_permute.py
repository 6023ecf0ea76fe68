"""MoE token permutation (reference: d9d/kernel/moe/permute_with_probs.py,
indices_to_multihot.py).

`moe_permute`: replicate each token once per selected expert and sort the
replicas by expert id, so each expert's rows are contiguous for the grouped
GEMM. `moe_unpermute`: sum a token's replicas back (prob-weighted).

v1 is index-op based (argsort + index_select / index_add — fully
differentiable, runs on GPU via ATen's HIP kernels); the dedicated CDNA4
gather/scatter kernels replace the hot paths next.
"""

import torch


def moe_permute(
    tokens: torch.Tensor,  # (T, H)
    expert_indices: torch.Tensor,  # (T, K) int64
    probs: torch.Tensor,  # (T, K)
    num_experts: int,
):
    """Returns (permuted_tokens (T*K, H), permuted_probs (T*K,),
    row_to_token (T*K,), tokens_per_expert (E,))."""
    T, K = expert_indices.shape
    flat_experts = expert_indices.reshape(-1)  # (T*K,)
    order = torch.argsort(flat_experts, stable=True)
    row_to_token = order // K  # source token per permuted row
    permuted_tokens = tokens.index_select(0, row_to_token)
    permuted_probs = probs.reshape(-1).index_select(0, order)
    tokens_per_expert = torch.bincount(flat_experts, minlength=num_experts)
    return permuted_tokens, permuted_probs, row_to_token, tokens_per_expert


def moe_unpermute(
    expert_out: torch.Tensor,  # (T*K, H)
    permuted_probs: torch.Tensor,  # (T*K,)
    row_to_token: torch.Tensor,  # (T*K,)
    num_tokens: int,
) -> torch.Tensor:
    """Weighted sum of each token's expert outputs back to (T, H)."""
    weighted = expert_out * permuted_probs.unsqueeze(-1).to(expert_out.dtype)
    out = torch.zeros(
        (num_tokens, expert_out.shape[1]), dtype=expert_out.dtype, device=expert_out.device
    )
    return out.index_add(0, row_to_token, weighted)
