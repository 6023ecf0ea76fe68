"""LoRA for nn.Linear and GroupedLinear (reference: d9d/peft/lora/layer.py:9,83)."""

import math
import re

import torch
from torch import nn

from ..module.block.moe.grouped_linear import GroupedLinear
from ..ops import gmm
from .base import PeftMethod


class LoRALinear(nn.Module):
    def __init__(self, base: nn.Linear, rank: int, alpha: float, dropout: float = 0.0):
        super().__init__()
        self.base = base
        self.rank = rank
        self.scaling = alpha / rank
        self.lora_A = nn.Parameter(
            torch.zeros(rank, base.in_features, device=base.weight.device,
                        dtype=base.weight.dtype)
        )
        self.lora_B = nn.Parameter(
            torch.zeros(base.out_features, rank, device=base.weight.device,
                        dtype=base.weight.dtype)
        )
        self.dropout = nn.Dropout(dropout)
        self.reset_parameters()

    def reset_parameters(self) -> None:
        with torch.no_grad():
            nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
            nn.init.zeros_(self.lora_B)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.base(x)
        lora = self.dropout(x) @ self.lora_A.t() @ self.lora_B.t()
        return out + lora * self.scaling

    @torch.no_grad()
    def merge_into_base(self) -> nn.Linear:
        self.base.weight += (self.lora_B @ self.lora_A) * self.scaling
        return self.base


class LoRAGroupedLinear(nn.Module):
    """3-D A/B adapters over grouped expert weights (reference: lora/layer.py:83)."""

    def __init__(self, base: GroupedLinear, rank: int, alpha: float):
        super().__init__()
        self.base = base
        self.rank = rank
        self.scaling = alpha / rank
        w = base.weight
        local = w.to_local() if hasattr(w, "to_local") else w
        E = local.shape[0]
        self.lora_A = nn.Parameter(
            torch.zeros(E, base.in_features, rank, device=local.device, dtype=local.dtype)
        )
        self.lora_B = nn.Parameter(
            torch.zeros(E, rank, base.out_features, device=local.device, dtype=local.dtype)
        )
        self.reset_parameters()

    def reset_parameters(self) -> None:
        with torch.no_grad():
            nn.init.normal_(self.lora_A, std=1.0 / math.sqrt(self.base.in_features))
            nn.init.zeros_(self.lora_B)

    def forward(self, x: torch.Tensor, batch_sizes: torch.Tensor) -> torch.Tensor:
        out = self.base(x, batch_sizes)
        lora = gmm(gmm(x, self.lora_A, batch_sizes), self.lora_B, batch_sizes)
        return out + lora * self.scaling

    @torch.no_grad()
    def merge_into_base(self) -> GroupedLinear:
        w = self.base.weight
        local = w.to_local() if hasattr(w, "to_local") else w
        # base weight is (E, out, in); A@B is (E, in, out)
        local += (torch.bmm(self.lora_A, self.lora_B) * self.scaling).transpose(1, 2)
        return self.base


class LoRAMethod(PeftMethod):
    def __init__(
        self,
        rank: int = 8,
        alpha: float = 16.0,
        dropout: float = 0.0,
        target_patterns: tuple[str, ...] = (r".*proj$",),
    ) -> None:
        self.rank = rank
        self.alpha = alpha
        self.dropout = dropout
        self.target_patterns = [re.compile(p) for p in target_patterns]

    def _matches(self, name: str) -> bool:
        return any(p.match(name) for p in self.target_patterns)

    def inject(self, module: nn.Module) -> nn.Module:
        for parent_name, parent in list(module.named_modules()):
            for child_name, child in list(parent.named_children()):
                fqn = f"{parent_name}.{child_name}" if parent_name else child_name
                if not self._matches(fqn):
                    continue
                if isinstance(child, nn.Linear):
                    setattr(parent, child_name,
                            LoRALinear(child, self.rank, self.alpha, self.dropout))
                elif isinstance(child, GroupedLinear):
                    setattr(parent, child_name,
                            LoRAGroupedLinear(child, self.rank, self.alpha))
        return module

    def merge(self, module: nn.Module) -> nn.Module:
        for parent_name, parent in list(module.named_modules()):
            for child_name, child in list(parent.named_children()):
                if isinstance(child, (LoRALinear, LoRAGroupedLinear)):
                    setattr(parent, child_name, child.merge_into_base())
        return module
