"""RoPE frequency scalings (reference: d9d/module/block/positional/rope_scaling.py).

Each scaling transforms the base inverse frequencies and optionally applies an
attention-magnitude scale (YaRN mscale).
"""

import math
from dataclasses import dataclass

import torch


@dataclass(frozen=True)
class RopeScaling:
    def scale_inv_freq(self, inv_freq: torch.Tensor) -> torch.Tensor:
        return inv_freq

    @property
    def mscale(self) -> float:
        return 1.0


@dataclass(frozen=True)
class NoScaling(RopeScaling):
    pass


@dataclass(frozen=True)
class LinearScaling(RopeScaling):
    factor: float = 1.0

    def scale_inv_freq(self, inv_freq: torch.Tensor) -> torch.Tensor:
        return inv_freq / self.factor


@dataclass(frozen=True)
class NTKScaling(RopeScaling):
    """NTK-aware base stretch: base' = base * factor^(dim/(dim-2))."""

    factor: float = 1.0
    rope_dim: int = 128

    def scale_inv_freq(self, inv_freq: torch.Tensor) -> torch.Tensor:
        # base' = base * factor^(d/(d-2)); inv_freq_i = base'^(-2i/d), so the
        # per-dim scale is factor^(-(2i/d) * d/(d-2)).
        d = self.rope_dim
        i2 = torch.arange(0, d, 2, device=inv_freq.device, dtype=inv_freq.dtype)
        return inv_freq * self.factor ** (-(i2 / d) * d / max(d - 2, 1))


@dataclass(frozen=True)
class YarnScaling(RopeScaling):
    """YaRN: interpolate low frequencies, keep high frequencies, mscale boost."""

    factor: float = 1.0
    original_max_position: int = 4096
    beta_fast: float = 32.0
    beta_slow: float = 1.0
    mscale_all_dim: float = 0.1

    def _ramp(self, inv_freq: torch.Tensor) -> torch.Tensor:
        # Per-dim interpolation weight in [0, 1]: 1 = fully interpolated (low freq).
        rotations = self.original_max_position * inv_freq / (2 * math.pi)
        low, high = self.beta_slow, self.beta_fast
        t = (rotations - low) / max(high - low, 1e-6)
        return 1.0 - t.clamp(0.0, 1.0)

    def scale_inv_freq(self, inv_freq: torch.Tensor) -> torch.Tensor:
        interp = inv_freq / self.factor
        mask = self._ramp(inv_freq)
        return interp * mask + inv_freq * (1.0 - mask)

    @property
    def mscale(self) -> float:
        if self.factor <= 1.0:
            return 1.0
        return 0.1 * self.mscale_all_dim * math.log(self.factor) + 1.0
