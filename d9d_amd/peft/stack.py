"""PEFT composition (reference: d9d/peft/all/method.py:14)."""

from torch import nn

from .base import PeftMethod


class PeftStack(PeftMethod):
    def __init__(self, *methods: PeftMethod) -> None:
        self.methods = list(methods)

    def inject(self, module: nn.Module) -> nn.Module:
        for m in self.methods:
            module = m.inject(module)
        return module

    def merge(self, module: nn.Module) -> nn.Module:
        for m in reversed(self.methods):
            module = m.merge(module)
        return module
