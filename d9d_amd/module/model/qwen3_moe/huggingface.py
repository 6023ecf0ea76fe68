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
    StateGroup,
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


class _FusedToStacked(ModelStateMapper):
    """HF packed experts -> our stacked (E, out, in) weights for one layer."""

    def __init__(self, prefix: str, intermediate: int) -> None:
        self.prefix = prefix
        self.intermediate = intermediate

    def state_dependency_groups(self):
        pre = self.prefix
        return [
            StateGroup.of(
                [pre + "mlp.experts.gate_up_proj"],
                [pre + "mlp.experts.gate_proj.weight", pre + "mlp.experts.up_proj.weight"],
            ),
            StateGroup.of(
                [pre + "mlp.experts.down_proj"],
                [pre + "mlp.experts.down_proj.weight"],
            ),
        ]

    def apply_group(self, group, tensors):
        pre = self.prefix
        if pre + "mlp.experts.gate_up_proj" in tensors:
            gu = tensors[pre + "mlp.experts.gate_up_proj"]  # (E, 2I, H)
            # our GroupedLinear stores (E, out, in) = (E, I, H): direct split
            gate = gu[:, : self.intermediate, :].contiguous()
            up = gu[:, self.intermediate :, :].contiguous()
            return {
                pre + "mlp.experts.gate_proj.weight": gate,
                pre + "mlp.experts.up_proj.weight": up,
            }
        down = tensors[pre + "mlp.experts.down_proj"]  # (E, H, I) = (E, out, in)
        return {pre + "mlp.experts.down_proj.weight": down.contiguous()}


class _StackedToFused(ModelStateMapper):
    def __init__(self, prefix: str, intermediate: int) -> None:
        self.prefix = prefix
        self.intermediate = intermediate

    def state_dependency_groups(self):
        pre = self.prefix
        return [
            StateGroup.of(
                [pre + "mlp.experts.gate_proj.weight", pre + "mlp.experts.up_proj.weight"],
                [pre + "mlp.experts.gate_up_proj"],
            ),
            StateGroup.of(
                [pre + "mlp.experts.down_proj.weight"],
                [pre + "mlp.experts.down_proj"],
            ),
        ]

    def apply_group(self, group, tensors):
        import torch

        pre = self.prefix
        if pre + "mlp.experts.gate_proj.weight" in tensors:
            gate = tensors[pre + "mlp.experts.gate_proj.weight"]
            up = tensors[pre + "mlp.experts.up_proj.weight"]
            return {pre + "mlp.experts.gate_up_proj": torch.cat([gate, up], dim=1).contiguous()}
        down = tensors[pre + "mlp.experts.down_proj.weight"]
        return {pre + "mlp.experts.down_proj": down.contiguous()}


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
            mappers.append(_FusedToStacked(pre, p.intermediate_size))
        else:
            for proj in ("gate_proj", "up_proj", "down_proj"):
                # HF expert weight (out, in) == our per-expert layout: stack
                mappers.append(
                    Sequential(
                        StackTensors(
                            [f"{pre}mlp.experts.{e}.{proj}.weight" for e in range(p.num_experts)],
                            f"{pre}mlp.experts.{proj}.weight",
                            dim=0,
                        ),
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
            mappers.append(_StackedToFused(pre, p.intermediate_size))
        else:
            for proj in ("gate_proj", "up_proj", "down_proj"):
                mappers.append(
                    UnstackTensors(
                        f"{pre}mlp.experts.{proj}.weight",
                        [f"{pre}mlp.experts.{e}.{proj}.weight" for e in range(p.num_experts)],
                        dim=0,
                    )
                )
    return Parallel(*mappers)
