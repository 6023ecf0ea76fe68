"""PEFT application: freeze base, inject adapters (reference: d9d/peft/applicator.py:9-36)."""

from torch import nn

from ..model_state import identity_mapper_from_module
from .base import PeftMethod


def inject_peft_and_freeze(module: nn.Module, method: PeftMethod):
    """Freeze everything, then let the method unfreeze/inject. Returns
    (module, state_mapper) for the transformed topology."""
    for p in module.parameters():
        p.requires_grad_(False)
    module = method.inject(module)
    mapper = identity_mapper_from_module(module)
    return module, mapper
