"""PEFT method ABC (reference: d9d/peft/base.py:27)."""

from abc import ABC, abstractmethod

from torch import nn


class PeftMethod(ABC):
    @abstractmethod
    def inject(self, module: nn.Module) -> nn.Module:
        """Transform the module in place (wrap layers, unfreeze params)."""

    @abstractmethod
    def merge(self, module: nn.Module) -> nn.Module:
        """Fold adapters back into base weights."""
