"""HF <-> d9d_amd state mappers for Qwen3 dense (reference: qwen3_dense/huggingface.py)."""

from ....model_state.mapper import (
    ConcatenateTensors,
    Identity,
    ModelStateMapper,
    Parallel,
    SliceRows,
)
from .params import Qwen3DenseModelParameters


def _vocab_splits(p: Qwen3DenseModelParameters) -> list[tuple[str, int]]:
    return [(name, p.split_vocab_size[name]) for name in p.split_vocab_order]


def _layer_keys(p: Qwen3DenseModelParameters) -> tuple[str, ...]:
    keys = [
        "self_attn.q_proj.weight",
        "self_attn.k_proj.weight",
        "self_attn.v_proj.weight",
        "self_attn.o_proj.weight",
    ]
    if p.use_qk_norm:
        keys += ["self_attn.q_norm.weight", "self_attn.k_norm.weight"]
    keys += [
        "input_layernorm.weight",
        "post_attention_layernorm.weight",
        "mlp.gate_proj.weight",
        "mlp.up_proj.weight",
        "mlp.down_proj.weight",
    ]
    return tuple(keys)


def hf_to_d9d_mapper(p: Qwen3DenseModelParameters) -> ModelStateMapper:
    mappers: list[ModelStateMapper] = [
        SliceRows(
            "model.embed_tokens.weight",
            [
                (f"model.embed_tokens.embeddings.{name}.weight", size)
                for name, size in _vocab_splits(p)
            ],
        ),
        SliceRows(
            "lm_head.weight",
            [(f"lm_head.weights.{name}", size) for name, size in _vocab_splits(p)],
        ),
        Identity("model.norm.weight"),
    ]
    for i in range(p.num_hidden_layers):
        for key in _layer_keys(p):
            mappers.append(Identity(f"model.layers.{i}.{key}"))
    return Parallel(*mappers)


def d9d_to_hf_mapper(p: Qwen3DenseModelParameters) -> ModelStateMapper:
    mappers: list[ModelStateMapper] = [
        ConcatenateTensors(
            [
                f"model.embed_tokens.embeddings.{name}.weight"
                for name, _ in _vocab_splits(p)
            ],
            "model.embed_tokens.weight",
        ),
        ConcatenateTensors(
            [f"lm_head.weights.{name}" for name, _ in _vocab_splits(p)],
            "lm_head.weight",
        ),
        Identity("model.norm.weight"),
    ]
    for i in range(p.num_hidden_layers):
        for key in _layer_keys(p):
            mappers.append(Identity(f"model.layers.{i}.{key}"))
    return Parallel(*mappers)
