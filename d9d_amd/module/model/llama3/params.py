"""Llama-3 model parameters.

Llama-3 is the Qwen3-dense architecture minus q/k head norms (and with its
own rope base / vocab): the decoder stack, GQA attention, SwiGLU FFN and
split-vocab embedding are shared with `qwen3_dense`. BASELINE.json names
Llama-3 70B TP+SP+PP as a target config; the reference itself ships only
the Qwen3 families, so this family is native to this framework.
"""

from dataclasses import dataclass, field

from ..qwen3_dense.params import Qwen3DenseModelParameters


@dataclass(frozen=True)
class Llama3ModelParameters(Qwen3DenseModelParameters):
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    head_dim: int = 128
    num_hidden_layers: int = 32
    use_qk_norm: bool = False
    rope_base: float = 500_000.0
    max_position_ids: int = 131_072
    split_vocab_size: dict = field(
        default_factory=lambda: {"regular": 128_000, "special": 256}
    )

    @staticmethod
    def tiny() -> "Llama3ModelParameters":
        return Llama3ModelParameters(
            hidden_size=64,
            intermediate_size=128,
            num_attention_heads=4,
            num_key_value_heads=2,
            head_dim=16,
            num_hidden_layers=2,
            split_vocab_size={"regular": 128, "special": 8},
        )

    @staticmethod
    def llama3_8b() -> "Llama3ModelParameters":
        return Llama3ModelParameters()

    @staticmethod
    def llama3_70b() -> "Llama3ModelParameters":
        return Llama3ModelParameters(
            hidden_size=8192,
            intermediate_size=28672,
            num_attention_heads=64,
            num_key_value_heads=8,
            num_hidden_layers=80,
        )
