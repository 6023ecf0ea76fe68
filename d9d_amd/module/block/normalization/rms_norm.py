"""RMSNorm module over the CDNA4 kernel (reference: d9d/module/block/normalization/rms_norm.py)."""

import torch
from torch import nn

from ....ops import rms_norm


class RMSNorm(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        eps: float = 1e-6,
        zero_centered: bool = False,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        self.hidden_size = hidden_size
        self.eps = eps
        self.zero_centered = zero_centered
        self.weight = nn.Parameter(
            torch.empty(hidden_size, device=device, dtype=dtype)
        )

    def reset_parameters(self) -> None:
        with torch.no_grad():
            if self.zero_centered:
                self.weight.zero_()
            else:
                self.weight.fill_(1.0)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape
        out = rms_norm(
            x.reshape(-1, shape[-1]), self.weight, self.eps, self.zero_centered
        )
        return out.reshape(shape)

    def extra_repr(self) -> str:
        return f"{self.hidden_size}, eps={self.eps}, zero_centered={self.zero_centered}"
