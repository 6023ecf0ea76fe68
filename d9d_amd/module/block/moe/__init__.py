from .router import TopKRouter
from .grouped_linear import GroupedLinear
from .grouped_experts import GroupedSwiGLU
from .shared_expert import SharedSwiGLU
from .layer import MoELayer

__all__ = ["TopKRouter", "GroupedLinear", "GroupedSwiGLU", "SharedSwiGLU", "MoELayer"]
