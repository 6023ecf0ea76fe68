from .params import Qwen3MoEModelParameters
from .model import (
    Qwen3MoEDecoderLayer,
    Qwen3MoEModel,
    Qwen3MoEForCausalLM,
)

__all__ = [
    "Qwen3MoEModelParameters",
    "Qwen3MoEDecoderLayer",
    "Qwen3MoEModel",
    "Qwen3MoEForCausalLM",
]
