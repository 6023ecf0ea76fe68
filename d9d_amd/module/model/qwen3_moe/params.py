"""Qwen3-MoE model parameters (reference: d9d/module/model/qwen3_moe params)."""

from dataclasses import dataclass, field


@dataclass(frozen=True)
class Qwen3MoEModelParameters:
    hidden_size: int = 768
    intermediate_size: int = 576  # per expert
    num_experts: int = 128
    experts_top_k: int = 8
    num_attention_heads: int = 16
    num_key_value_heads: int = 4
    head_dim: int = 128
    num_hidden_layers: int = 16
    rms_norm_eps: float = 1e-6
    rope_base: float = 1_000_000.0
    max_position_ids: int = 128_000
    split_vocab_size: dict = field(
        default_factory=lambda: {"regular": 151_643, "special": 26}
    )
    split_vocab_order: tuple = ("regular", "special")
    shared_expert_intermediate_size: int | None = None
    use_expert_bias: bool = False
    pipeline_num_virtual_layers_pre: int = 0
    pipeline_num_virtual_layers_post: int = 1
    checkpoint_layers: bool = False

    @property
    def vocab_size(self) -> int:
        return sum(self.split_vocab_size.values())

    @staticmethod
    def tiny() -> "Qwen3MoEModelParameters":
        return Qwen3MoEModelParameters(
            hidden_size=64,
            intermediate_size=48,
            num_experts=8,
            experts_top_k=2,
            num_attention_heads=4,
            num_key_value_heads=2,
            head_dim=16,
            num_hidden_layers=2,
            split_vocab_size={"regular": 500, "special": 12},
        )

    @staticmethod
    def example_pretrain() -> "Qwen3MoEModelParameters":
        """The reference example config (example/qwen3_moe/pretrain.json)."""
        return Qwen3MoEModelParameters()

    @staticmethod
    def qwen3_30b_a3b() -> "Qwen3MoEModelParameters":
        """Qwen3-30B-A3B (BASELINE.json config: MoE pretrain, EP=8 over xGMI)."""
        return Qwen3MoEModelParameters(
            hidden_size=2048,
            intermediate_size=768,
            num_experts=128,
            experts_top_k=8,
            num_attention_heads=32,
            num_key_value_heads=4,
            head_dim=128,
            num_hidden_layers=48,
            split_vocab_size={"regular": 151_643, "special": 26},
        )
