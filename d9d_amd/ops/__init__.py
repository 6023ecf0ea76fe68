from .rms_norm import rms_norm
from .swiglu import silu_mul, silu_mul_packed
from .stochastic import copy_fp32_to_bf16_stochastic_, adamw_stochastic_bf16_
from .attention import flash_attn_func, flash_attn_varlen_func
from .cce import linear_cross_entropy, LM_IGNORE_INDEX, VocabParallelOptions
from .gmm import gmm, gmm_nt
from .moe_permute import moe_permute, moe_unpermute
from .router import router_topk

__all__ = [
    "rms_norm",
    "router_topk",
    "silu_mul",
    "silu_mul_packed",
    "copy_fp32_to_bf16_stochastic_",
    "adamw_stochastic_bf16_",
    "flash_attn_func",
    "flash_attn_varlen_func",
    "linear_cross_entropy",
    "LM_IGNORE_INDEX",
    "VocabParallelOptions",
    "gmm",
    "gmm_nt",
    "moe_permute",
    "moe_unpermute",
]
