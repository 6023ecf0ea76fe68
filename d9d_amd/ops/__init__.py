from .rms_norm import rms_norm
from .swiglu import silu_mul
from .stochastic import copy_fp32_to_bf16_stochastic_, adamw_stochastic_bf16_

__all__ = [
    "rms_norm",
    "silu_mul",
    "copy_fp32_to_bf16_stochastic_",
    "adamw_stochastic_bf16_",
]
