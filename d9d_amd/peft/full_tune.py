"""Full-tune method: regex unfreeze (reference: d9d/peft/full/)."""

import re

from torch import nn

from .base import PeftMethod


class FullTuneMethod(PeftMethod):
    def __init__(self, patterns: tuple[str, ...] = (r".*",)) -> None:
        self.patterns = [re.compile(p) for p in patterns]

    def inject(self, module: nn.Module) -> nn.Module:
        for name, p in module.named_parameters():
            if any(pat.match(name) for pat in self.patterns):
                p.requires_grad_(True)
        return module

    def merge(self, module: nn.Module) -> nn.Module:
        return module
