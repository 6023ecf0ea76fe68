from .base import PeftMethod
from .lora import LoRAMethod, LoRALinear, LoRAGroupedLinear
from .full_tune import FullTuneMethod
from .stack import PeftStack
from .applicator import inject_peft_and_freeze

__all__ = [
    "PeftMethod",
    "LoRAMethod",
    "LoRALinear",
    "LoRAGroupedLinear",
    "FullTuneMethod",
    "PeftStack",
    "inject_peft_and_freeze",
]
