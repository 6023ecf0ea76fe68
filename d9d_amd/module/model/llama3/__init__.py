from ..qwen3_dense.huggingface import d9d_to_hf_mapper, hf_to_d9d_mapper
from ..qwen3_dense.model import Qwen3DenseForCausalLM as Llama3ForCausalLM
from ..qwen3_dense.model import Qwen3DenseModel as Llama3Model
from .params import Llama3ModelParameters

__all__ = [
    "Llama3Model",
    "Llama3ForCausalLM",
    "Llama3ModelParameters",
    "hf_to_d9d_mapper",
    "d9d_to_hf_mapper",
]
