"""Grouped-query attention block (reference: d9d/module/block/attention/grouped_query.py).

q/k/v/o projections, optional QK-RMSNorm (optionally zero-centered), partial
RoPE (`rope_dim` <= head_dim), optional sigmoid output gate (Qwen3.5 style),
SDPA = the framework's CDNA4 flash-attention op.
"""

import math

import torch
from torch import nn

from ..linear import KernelLinear

from ....ops import flash_attn_func
from ..normalization import RMSNorm
from ..positional import RopeLayout, apply_rotary_emb


class GroupedQueryAttention(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        num_attention_heads: int,
        num_key_value_heads: int,
        head_dim: int,
        rope_dim: int | None = None,
        rope_layout: RopeLayout = RopeLayout.HALF,
        use_qk_norm: bool = True,
        qk_norm_zero_centered: bool = False,
        rms_norm_eps: float = 1e-6,
        use_output_gate: bool = False,
        sliding_window: int | None = None,
        use_sinks: bool = False,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        assert num_attention_heads % num_key_value_heads == 0
        self.num_heads = num_attention_heads
        self.num_kv_heads = num_key_value_heads
        self.head_dim = head_dim
        self.rope_dim = rope_dim if rope_dim is not None else head_dim
        self.rope_layout = rope_layout
        self.sliding_window = sliding_window

        kw = {"device": device, "dtype": dtype, "bias": False}
        q_out = num_attention_heads * head_dim
        self.q_proj = KernelLinear(hidden_size, q_out * (2 if use_output_gate else 1), **kw)
        self.k_proj = KernelLinear(hidden_size, num_key_value_heads * head_dim, **kw)
        self.v_proj = KernelLinear(hidden_size, num_key_value_heads * head_dim, **kw)
        self.o_proj = KernelLinear(q_out, hidden_size, **kw)
        self.use_output_gate = use_output_gate

        if use_qk_norm:
            self.q_norm = RMSNorm(head_dim, eps=rms_norm_eps,
                                  zero_centered=qk_norm_zero_centered,
                                  device=device, dtype=dtype)
            self.k_norm = RMSNorm(head_dim, eps=rms_norm_eps,
                                  zero_centered=qk_norm_zero_centered,
                                  device=device, dtype=dtype)
        else:
            self.q_norm = None
            self.k_norm = None

        if use_sinks:
            self.sinks = nn.Parameter(
                torch.empty(num_attention_heads, device=device, dtype=torch.float32)
            )
        else:
            self.sinks = None

        # Context parallelism: installed by parallelize_context_parallel.
        # When set, K/V are all-gathered along the sequence over the cp group
        # and the local Q block attends with its global position offset.
        self._cp_group = None
        self._cp_rank = 0
        self._cp_size = 1
        self._cp_mode = "ring"

    def reset_parameters(self) -> None:
        with torch.no_grad():
            for lin in (self.q_proj, self.k_proj, self.v_proj, self.o_proj):
                nn.init.normal_(lin.weight, mean=0.0, std=0.02 / math.sqrt(2))
            if self.q_norm is not None:
                self.q_norm.reset_parameters()
                self.k_norm.reset_parameters()
            if self.sinks is not None:
                nn.init.zeros_(self.sinks)

    def forward(
        self,
        hidden_states: torch.Tensor,  # (B, S, H)
        rotary_cos_sin: tuple[torch.Tensor, torch.Tensor],
        cu_seqlens: torch.Tensor | None = None,  # packed documents (B == 1)
    ) -> torch.Tensor:
        B, S, _ = hidden_states.shape
        q = self.q_proj(hidden_states)
        if self.use_output_gate:
            q, gate = q.chunk(2, dim=-1)
        k = self.k_proj(hidden_states).view(B, S, self.num_kv_heads, self.head_dim)
        v = self.v_proj(hidden_states).view(B, S, self.num_kv_heads, self.head_dim)
        q = q.view(B, S, self.num_heads, self.head_dim)

        if self.q_norm is not None:
            q = self.q_norm(q)
            k = self.k_norm(k)

        cos, sin = rotary_cos_sin
        cos = cos[..., : self.rope_dim]
        sin = sin[..., : self.rope_dim]
        from ....ops.rope import rope_qk, rope_qk_available

        if self.rope_layout is RopeLayout.HALF and rope_qk_available(q, self.rope_dim):
            q, k = rope_qk(q, k, cos, sin)
        else:
            q = apply_rotary_emb(q, cos, sin, self.rope_layout)
            k = apply_rotary_emb(k, cos, sin, self.rope_layout)

        window = (-1, -1) if self.sliding_window is None else (self.sliding_window, -1)
        if cu_seqlens is not None:
            # packed documents: attention is causal WITHIN each document
            # (varlen path; rotary positions must already restart per doc)
            assert B == 1, "packed cu_seqlens input expects batch dim 1"
            from ....ops import flash_attn_varlen_func

            attn = flash_attn_varlen_func(
                q.squeeze(0), k.squeeze(0), v.squeeze(0), cu_seqlens,
                causal=True, window_size=window, sinks=self.sinks,
            ).unsqueeze(0)
            attn = attn.reshape(B, S, self.num_heads * self.head_dim)
            if self.use_output_gate:
                attn = attn * torch.sigmoid(gate)
            return self.o_proj(attn)
        q_offset = 0
        if self._cp_group is not None and self._cp_size > 1:
            if getattr(self, "_cp_mode", "ring") == "ring":
                # ring attention: KV chunks travel the ring, partials merge
                # by LSE — O(S/cp) resident memory
                assert self.sinks is None and self.sliding_window is None, (
                    "ring CP supports plain causal attention (no sinks/window)"
                )
                from ....parallel.context import ring_attention

                attn = ring_attention(q, k, v, group=self._cp_group, causal=True)
                attn = attn.reshape(B, S, self.num_heads * self.head_dim)
                if self.use_output_gate:
                    attn = attn * torch.sigmoid(gate)
                return self.o_proj(attn)
            from ....parallel.tensor import _AllGatherSeq

            k = _AllGatherSeq.apply(k, self._cp_group, 1)
            v = _AllGatherSeq.apply(v, self._cp_group, 1)
            q_offset = self._cp_rank * S
        attn = flash_attn_func(
            q, k, v, causal=True, window_size=window, sinks=self.sinks,
            q_offset=q_offset,
        )
        attn = attn.reshape(B, S, self.num_heads * self.head_dim)
        if self.use_output_gate:
            attn = attn * torch.sigmoid(gate)
        return self.o_proj(attn)
