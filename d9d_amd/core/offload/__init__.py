from .api import SleepTag, Offloadable
from .tensor import offload_tensor, onload_tensor

__all__ = ["SleepTag", "Offloadable", "offload_tensor", "onload_tensor"]
