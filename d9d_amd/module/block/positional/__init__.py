from .rope import (
    RopeLayout,
    RotaryEmbeddingProvider,
    apply_rotary_emb,
)
from .rope_scaling import (
    RopeScaling,
    NoScaling,
    LinearScaling,
    NTKScaling,
    YarnScaling,
)

__all__ = [
    "RopeLayout",
    "RotaryEmbeddingProvider",
    "apply_rotary_emb",
    "RopeScaling",
    "NoScaling",
    "LinearScaling",
    "NTKScaling",
    "YarnScaling",
]
