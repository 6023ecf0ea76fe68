from .tensor import (
    gather,
    all_gather,
    all_gather_variadic_shape,
    gather_variadic_shape,
)
from .object import gather_object, all_gather_object

__all__ = [
    "gather",
    "all_gather",
    "all_gather_variadic_shape",
    "gather_variadic_shape",
    "gather_object",
    "all_gather_object",
]
