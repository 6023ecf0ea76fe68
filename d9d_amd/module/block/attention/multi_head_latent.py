"""Multi-head latent attention (DeepSeek-V2 MLA).

Reference: d9d/module/block/attention/multi_head_latent.py — optional
low-rank Q projection, KV down-projection to `kv_lora_rank` plus a shared
rope sub-vector, RMSNorm on the latents, up-projection to per-head
(qk_nope + v); V is zero-padded to the qk head dim so the flash kernel's
uniform head_dim applies (reference notes this at :197).
"""

import math

import torch
from torch import nn

from ....ops import flash_attn_func
from ..normalization import RMSNorm
from ..positional import apply_rotary_emb


class LowRankProjection(nn.Module):
    """x -> down (rank) -> norm -> up (out)."""

    def __init__(self, in_features: int, rank: int, out_features: int,
                 eps: float = 1e-6, device=None, dtype=None) -> None:
        super().__init__()
        kw = {"device": device, "dtype": dtype, "bias": False}
        self.down = nn.Linear(in_features, rank, **kw)
        self.norm = RMSNorm(rank, eps=eps, device=device, dtype=dtype)
        self.up = nn.Linear(rank, out_features, **kw)

    def reset_parameters(self) -> None:
        with torch.no_grad():
            nn.init.normal_(self.down.weight, std=0.02)
            nn.init.normal_(self.up.weight, std=0.02)
        self.norm.reset_parameters()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.up(self.norm(self.down(x)))


class MultiHeadLatentAttention(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        num_attention_heads: int,
        qk_nope_head_dim: int = 128,
        qk_rope_head_dim: int = 64,
        v_head_dim: int = 128,
        kv_lora_rank: int = 512,
        q_lora_rank: int | None = None,
        rms_norm_eps: float = 1e-6,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        kw = {"device": device, "dtype": dtype, "bias": False}
        self.num_heads = num_attention_heads
        self.qk_nope_head_dim = qk_nope_head_dim
        self.qk_rope_head_dim = qk_rope_head_dim
        self.qk_head_dim = qk_nope_head_dim + qk_rope_head_dim
        self.v_head_dim = v_head_dim

        q_out = num_attention_heads * self.qk_head_dim
        if q_lora_rank is not None:
            self.q_proj = LowRankProjection(
                hidden_size, q_lora_rank, q_out, eps=rms_norm_eps,
                device=device, dtype=dtype,
            )
        else:
            self.q_proj = nn.Linear(hidden_size, q_out, **kw)

        # KV: latent (kv_lora_rank) + shared rope sub-vector per token
        self.kv_down = nn.Linear(hidden_size, kv_lora_rank + qk_rope_head_dim, **kw)
        self.kv_norm = RMSNorm(kv_lora_rank, eps=rms_norm_eps, device=device, dtype=dtype)
        self.kv_up = nn.Linear(
            kv_lora_rank,
            num_attention_heads * (qk_nope_head_dim + v_head_dim),
            **kw,
        )
        self.kv_lora_rank = kv_lora_rank
        self.o_proj = nn.Linear(num_attention_heads * v_head_dim, hidden_size, **kw)

    def reset_parameters(self) -> None:
        with torch.no_grad():
            for m in (self.kv_down, self.kv_up, self.o_proj):
                nn.init.normal_(m.weight, std=0.02 / math.sqrt(2))
            if isinstance(self.q_proj, nn.Linear):
                nn.init.normal_(self.q_proj.weight, std=0.02 / math.sqrt(2))
        if isinstance(self.q_proj, LowRankProjection):
            self.q_proj.reset_parameters()
        self.kv_norm.reset_parameters()

    def forward(
        self,
        hidden_states: torch.Tensor,  # (B, S, H)
        rotary_cos_sin: tuple[torch.Tensor, torch.Tensor],
    ) -> torch.Tensor:
        B, S, _ = hidden_states.shape
        H = self.num_heads

        q = self.q_proj(hidden_states).view(B, S, H, self.qk_head_dim)
        q_nope, q_rope = q.split([self.qk_nope_head_dim, self.qk_rope_head_dim], dim=-1)

        kv = self.kv_down(hidden_states)
        latent, k_rope = kv.split([self.kv_lora_rank, self.qk_rope_head_dim], dim=-1)
        latent = self.kv_norm(latent)
        kv_up = self.kv_up(latent).view(
            B, S, H, self.qk_nope_head_dim + self.v_head_dim
        )
        k_nope, v = kv_up.split([self.qk_nope_head_dim, self.v_head_dim], dim=-1)

        cos, sin = rotary_cos_sin
        cos = cos[..., : self.qk_rope_head_dim]
        sin = sin[..., : self.qk_rope_head_dim]
        q_rope = apply_rotary_emb(q_rope, cos, sin)
        # shared rope key: one per token, broadcast over heads
        k_rope = apply_rotary_emb(k_rope.unsqueeze(2), cos, sin).expand(B, S, H, -1)

        q_full = torch.cat([q_nope, q_rope], dim=-1)
        k_full = torch.cat([k_nope, k_rope], dim=-1)
        # pad V to qk_head_dim so one flash call handles both
        if self.v_head_dim < self.qk_head_dim:
            v_pad = torch.zeros(
                B, S, H, self.qk_head_dim - self.v_head_dim,
                device=v.device, dtype=v.dtype,
            )
            v_in = torch.cat([v, v_pad], dim=-1)
        else:
            v_in = v
        attn = flash_attn_func(
            q_full, k_full.contiguous(), v_in.contiguous(), causal=True,
            softmax_scale=1.0 / math.sqrt(self.qk_head_dim),
        )
        attn = attn[..., : self.v_head_dim].reshape(B, S, H * self.v_head_dim)
        return self.o_proj(attn)
