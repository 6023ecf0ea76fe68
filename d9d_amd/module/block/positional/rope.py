"""Rotary position embeddings (reference: d9d/module/block/positional/rope.py).

`RotaryEmbeddingProvider` buffers inverse frequencies and produces per-position
(cos, sin) tables; `apply_rotary_emb` rotates q/k. Two layouts:
HALF (rotate_half, HF/Llama convention) and INTERLEAVED (rotate_every_two,
GPT-NeoX native). Trig tables are computed once per forward from position_ids
(host-precomputed tables are the CDNA4 idiom: no per-element trig on device).
"""

import enum

import torch
from torch import nn

from .rope_scaling import NoScaling, RopeScaling


class RopeLayout(enum.Enum):
    HALF = "half"
    INTERLEAVED = "interleaved"


class RotaryEmbeddingProvider(nn.Module):
    """Produces (cos, sin) of shape (..., S, rope_dim) for given position ids."""

    inv_freq: torch.Tensor

    def __init__(
        self,
        rope_dim: int,
        base: float = 10000.0,
        scaling: RopeScaling | None = None,
        device=None,
    ) -> None:
        super().__init__()
        self.rope_dim = rope_dim
        self.base = base
        self.scaling = scaling or NoScaling()
        inv_freq = self._compute_inv_freq(device)
        self.register_buffer("inv_freq", inv_freq, persistent=False)

    def _compute_inv_freq(self, device) -> torch.Tensor:
        half = self.rope_dim // 2
        inv_freq = 1.0 / (
            self.base
            ** (torch.arange(0, half, device=device, dtype=torch.float32) * 2 / self.rope_dim)
        )
        return self.scaling.scale_inv_freq(inv_freq)

    def reset_parameters(self) -> None:
        with torch.no_grad():
            self.inv_freq.copy_(self._compute_inv_freq(self.inv_freq.device))

    def forward(self, position_ids: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        """position_ids (..., S) int -> cos/sin (..., S, rope_dim) fp32."""
        angles = position_ids.to(torch.float32).unsqueeze(-1) * self.inv_freq
        angles = torch.cat([angles, angles], dim=-1)  # HALF layout duplication
        mscale = self.scaling.mscale
        return torch.cos(angles) * mscale, torch.sin(angles) * mscale


def _rotate_half(x: torch.Tensor) -> torch.Tensor:
    half = x.shape[-1] // 2
    return torch.cat([-x[..., half:], x[..., :half]], dim=-1)


def _rotate_every_two(x: torch.Tensor) -> torch.Tensor:
    x1 = x[..., 0::2]
    x2 = x[..., 1::2]
    return torch.stack([-x2, x1], dim=-1).flatten(-2)


def apply_rotary_emb(
    x: torch.Tensor,  # (B, S, H, D_head)
    cos: torch.Tensor,  # (B, S, rope_dim) or (S, rope_dim)
    sin: torch.Tensor,
    layout: RopeLayout = RopeLayout.HALF,
) -> torch.Tensor:
    """Rotate the first rope_dim dims of x; pass the rest through (partial RoPE)."""
    rope_dim = cos.shape[-1]
    x_rope, x_pass = x[..., :rope_dim], x[..., rope_dim:]
    cos = cos.unsqueeze(-2).to(x.dtype)  # broadcast over heads
    sin = sin.unsqueeze(-2).to(x.dtype)
    if layout is RopeLayout.HALF:
        rotated = x_rope * cos + _rotate_half(x_rope) * sin
    else:
        # INTERLEAVED cos/sin come duplicated [f, f]; regroup as every-two.
        half = rope_dim // 2
        cos_i = cos[..., :half].repeat_interleave(2, dim=-1)
        sin_i = sin[..., :half].repeat_interleave(2, dim=-1)
        rotated = x_rope * cos_i + _rotate_every_two(x_rope) * sin_i
    if x_pass.shape[-1] == 0:
        return rotated
    return torch.cat([rotated, x_pass], dim=-1)
