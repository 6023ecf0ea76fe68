"""HF <-> d9d_amd state mappers for Qwen3-MoE (reference: qwen3_moe/huggingface.py).

Two HF expert formats are supported (reference: huggingface.py MODULE_LIST and
FUSED formats):
  * "module_list": model.layers.N.mlp.experts.M.{gate,up,down}_proj.weight,
    each (out, in) -> stack into our (E, out, in) (same per-expert layout).
  * "fused" (transformers >= 5.x packed experts):
    experts.gate_up_proj (E, 2I, H) and experts.down_proj (E, H, I)
    -> chunk gate/up on dim 1, transpose (1, 2).

Common differences from HF:
  * split vocab: embed_tokens.embeddings.{seg}.weight / lm_head.weights.{seg}
  * router: mlp.router.gate.weight (HF: mlp.gate.weight)
"""

from ....model_state.mapper import (
    ConcatenateTensors,
    Identity,
    ModelStateMapper,
    Parallel,
    Rename,
    Sequential,
    SliceRows,
    StackTensors,
    UnstackTensors,
)
from .params import Qwen3MoEModelParameters


def _vocab_splits(p: Qwen3MoEModelParameters) -> list[tuple[str, int]]:
    return [(name, p.split_vocab_size[name]) for name in p.split_vocab_order]


def hf_to_d9d_mapper(
    p: Qwen3MoEModelParameters, expert_format: str = "fused"
) -> ModelStateMapper:
    mappers: list[ModelStateMapper] = []

    mappers.append(
        SliceRows(
            "model.embed_tokens.weight",
            [
                (f"model.embed_tokens.embeddings.{name}.weight", size)
                for name, size in _vocab_splits(p)
            ],
        )
    )
    mappers.append(
        SliceRows(
            "lm_head.weight",
            [(f"lm_head.weights.{name}", size) for name, size in _vocab_splits(p)],
        )
    )
    mappers.append(Identity("model.norm.weight"))

    for i in range(p.num_hidden_layers):
        pre = f"model.layers.{i}."
        for key in (
            "self_attn.q_proj.weight",
            "self_attn.k_proj.weight",
            "self_attn.v_proj.weight",
            "self_attn.o_proj.weight",
            "self_attn.q_norm.weight",
            "self_attn.k_norm.weight",
            "input_layernorm.weight",
            "post_attention_layernorm.weight",
        ):
            mappers.append(Identity(pre + key))
        mappers.append(Rename(pre + "mlp.gate.weight", pre + "mlp.router.gate.weight"))
        if expert_format == "fused":
            # HF fused layout (E, 2I, H) / (E, H, I) == ours: pure renames
            mappers.append(
                Rename(pre + "mlp.experts.gate_up_proj", pre + "mlp.experts.gate_up_proj.weight")
            )
            mappers.append(
                Rename(pre + "mlp.experts.down_proj", pre + "mlp.experts.down_proj.weight")
            )
        else:
            # per-expert (out, in) weights: stack, then concat gate|up on dim 1
            for proj in ("gate_proj", "up_proj"):
                mappers.append(
                    StackTensors(
                        [f"{pre}mlp.experts.{e}.{proj}.weight" for e in range(p.num_experts)],
                        f"{pre}mlp.experts._stacked.{proj}",
                        dim=0,
                    )
                )
            mappers.append(
                Sequential(
                    ConcatenateTensors(
                        [pre + "mlp.experts._stacked.gate_proj", pre + "mlp.experts._stacked.up_proj"],
                        pre + "mlp.experts.gate_up_proj.weight",
                        dim=1,
                    ),
                )
            )
            mappers.append(
                StackTensors(
                    [f"{pre}mlp.experts.{e}.down_proj.weight" for e in range(p.num_experts)],
                    pre + "mlp.experts.down_proj.weight",
                    dim=0,
                )
            )
    return Parallel(*mappers)


def d9d_to_hf_mapper(
    p: Qwen3MoEModelParameters, expert_format: str = "fused"
) -> ModelStateMapper:
    mappers: list[ModelStateMapper] = []

    mappers.append(
        ConcatenateTensors(
            [
                f"model.embed_tokens.embeddings.{name}.weight"
                for name, _ in _vocab_splits(p)
            ],
            "model.embed_tokens.weight",
        )
    )
    mappers.append(
        ConcatenateTensors(
            [f"lm_head.weights.{name}" for name, _ in _vocab_splits(p)],
            "lm_head.weight",
        )
    )
    mappers.append(Identity("model.norm.weight"))

    for i in range(p.num_hidden_layers):
        pre = f"model.layers.{i}."
        for key in (
            "self_attn.q_proj.weight",
            "self_attn.k_proj.weight",
            "self_attn.v_proj.weight",
            "self_attn.o_proj.weight",
            "self_attn.q_norm.weight",
            "self_attn.k_norm.weight",
            "input_layernorm.weight",
            "post_attention_layernorm.weight",
        ):
            mappers.append(Identity(pre + key))
        mappers.append(Rename(pre + "mlp.router.gate.weight", pre + "mlp.gate.weight"))
        if expert_format == "fused":
            mappers.append(
                Rename(pre + "mlp.experts.gate_up_proj.weight", pre + "mlp.experts.gate_up_proj")
            )
            mappers.append(
                Rename(pre + "mlp.experts.down_proj.weight", pre + "mlp.experts.down_proj")
            )
        else:
            mappers.append(
                Sequential(
                    SliceRows(
                        pre + "mlp.experts.gate_up_proj.weight",
                        [
                            (pre + "mlp.experts._stacked.gate_proj", p.intermediate_size),
                            (pre + "mlp.experts._stacked.up_proj", p.intermediate_size),
                        ],
                        dim=1,
                    ),
                )
            )
            for proj in ("gate_proj", "up_proj"):
                mappers.append(
                    UnstackTensors(
                        f"{pre}mlp.experts._stacked.{proj}",
                        [f"{pre}mlp.experts.{e}.{proj}.weight" for e in range(p.num_experts)],
                        dim=0,
                    )
                )
            mappers.append(
                UnstackTensors(
                    pre + "mlp.experts.down_proj.weight",
                    [f"{pre}mlp.experts.{e}.down_proj.weight" for e in range(p.num_experts)],
                    dim=0,
                )
            )
    return Parallel(*mappers)
