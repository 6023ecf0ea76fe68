"""Qwen3 dense model parameters (reference: d9d/module/model/qwen3_dense)."""

from dataclasses import dataclass, field


@dataclass(frozen=True)
class Qwen3DenseModelParameters:
    hidden_size: int = 1024
    intermediate_size: int = 3072
    num_attention_heads: int = 16
    num_key_value_heads: int = 8
    head_dim: int = 128
    num_hidden_layers: int = 28
    rms_norm_eps: float = 1e-6
    use_qk_norm: bool = True
    rope_base: float = 1_000_000.0
    max_position_ids: int = 40_960
    split_vocab_size: dict = field(
        default_factory=lambda: {"regular": 151_643, "special": 26}
    )
    split_vocab_order: tuple = ("regular", "special")
    pipeline_num_virtual_layers_pre: int = 0
    pipeline_num_virtual_layers_post: int = 1
    checkpoint_layers: bool = False
    num_classes: int = 2  # classification head
    embedding_size: int | None = None  # embedding head projection

    @property
    def vocab_size(self) -> int:
        return sum(self.split_vocab_size.values())

    @staticmethod
    def tiny() -> "Qwen3DenseModelParameters":
        return Qwen3DenseModelParameters(
            hidden_size=64,
            intermediate_size=128,
            num_attention_heads=4,
            num_key_value_heads=2,
            head_dim=16,
            num_hidden_layers=2,
            split_vocab_size={"regular": 500, "special": 12},
        )

    @staticmethod
    def qwen3_0_6b() -> "Qwen3DenseModelParameters":
        """Qwen3-0.6B shape (BASELINE.json config #2)."""
        return Qwen3DenseModelParameters(
            hidden_size=1024,
            intermediate_size=3072,
            num_attention_heads=16,
            num_key_value_heads=8,
            head_dim=128,
            num_hidden_layers=28,
        )
