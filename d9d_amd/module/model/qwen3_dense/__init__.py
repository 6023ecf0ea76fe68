from .params import Qwen3DenseModelParameters
from .model import (
    Qwen3DenseDecoderLayer,
    Qwen3DenseModel,
    Qwen3DenseForCausalLM,
    Qwen3DenseForClassification,
    Qwen3DenseForEmbedding,
)

__all__ = [
    "Qwen3DenseModelParameters",
    "Qwen3DenseDecoderLayer",
    "Qwen3DenseModel",
    "Qwen3DenseForCausalLM",
    "Qwen3DenseForClassification",
    "Qwen3DenseForEmbedding",
]
