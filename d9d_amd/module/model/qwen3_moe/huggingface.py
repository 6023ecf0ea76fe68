"""HF <-> d9d_amd state mappers for Qwen3-MoE (reference: qwen3_moe/huggingface.py).

HF layout (transformers Qwen3MoeForCausalLM):
  model.embed_tokens.weight (V, H)
  model.layers.N.self_attn.{q,k,v,o}_proj.weight, {q,k}_norm.weight
  model.layers.N.{input,post_attention}_layernorm.weight
  model.layers.N.mlp.gate.weight (E, H)
  model.layers.N.mlp.experts.M.{gate,up,down}_proj.weight  (MODULE_LIST format)
  model.norm.weight, lm_head.weight (V, H)

d9d_amd layout differences:
  * split vocab: embed_tokens.embeddings.{seg}.weight / lm_head.weights.{seg}
  * stacked experts: mlp.experts.{gate,up,down}_proj.weight (E, in, out)
    (HF per-expert weights are (out, in) -> transpose then stack)
  * router: mlp.router.gate.weight
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
    Transpose,
    UnstackTensors,
)
from .params import Qwen3MoEModelParameters


def _vocab_splits(p: Qwen3MoEModelParameters) -> list[tuple[str, int]]:
    return [(name, p.split_vocab_size[name]) for name in p.split_vocab_order]


def hf_to_d9d_mapper(p: Qwen3MoEModelParameters) -> ModelStateMapper:
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
        for proj in ("gate_proj", "up_proj", "down_proj"):
            # HF expert weight (out, in) -> transpose (in, out) -> stack (E, in, out)
            per_expert = [
                Transpose(
                    f"{pre}mlp.experts.{e}.{proj}.weight",
                    f"{pre}mlp.experts._t{e}.{proj}",
                )
                for e in range(p.num_experts)
            ]
            mappers.append(
                Sequential(
                    Parallel(*per_expert),
                    StackTensors(
                        [f"{pre}mlp.experts._t{e}.{proj}" for e in range(p.num_experts)],
                        f"{pre}mlp.experts.{proj}.weight",
                        dim=0,
                    ),
                )
            )
    return Parallel(*mappers)


def d9d_to_hf_mapper(p: Qwen3MoEModelParameters) -> ModelStateMapper:
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
        for proj in ("gate_proj", "up_proj", "down_proj"):
            unstack = UnstackTensors(
                f"{pre}mlp.experts.{proj}.weight",
                [f"{pre}mlp.experts._t{e}.{proj}" for e in range(p.num_experts)],
                dim=0,
            )
            per_expert = [
                Transpose(
                    f"{pre}mlp.experts._t{e}.{proj}",
                    f"{pre}mlp.experts.{e}.{proj}.weight",
                )
                for e in range(p.num_experts)
            ]
            mappers.append(Sequential(unstack, Parallel(*per_expert)))
    return Parallel(*mappers)
