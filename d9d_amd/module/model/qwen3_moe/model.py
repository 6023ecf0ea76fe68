"""Qwen3-MoE model family (reference: d9d/module/model/qwen3_moe/model.py).

The model holds an nn.ModuleDict of GLOBALLY-indexed decoder layers so a
pipeline stage owns exactly its slice; embedding lives on the first stage and
norm+head on the last. Implements the pipelining shape-inference protocol.
"""

from typing import Any

import torch
from torch import nn
from torch.utils.checkpoint import checkpoint

from ....ops import LM_IGNORE_INDEX
from ....pipelining import PipelineStageInfo
from ...block.attention import GroupedQueryAttention
from ...block.embedding import SplitTokenEmbeddings
from ...block.head import SplitLanguageModellingHead
from ...block.moe import MoELayer
from ...block.normalization import RMSNorm
from ...block.positional import RotaryEmbeddingProvider
from .params import Qwen3MoEModelParameters


class Qwen3MoEDecoderLayer(nn.Module):
    def __init__(self, p: Qwen3MoEModelParameters, device=None, dtype=None) -> None:
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.input_layernorm = RMSNorm(p.hidden_size, eps=p.rms_norm_eps, **kw)
        self.self_attn = GroupedQueryAttention(
            hidden_size=p.hidden_size,
            num_attention_heads=p.num_attention_heads,
            num_key_value_heads=p.num_key_value_heads,
            head_dim=p.head_dim,
            rms_norm_eps=p.rms_norm_eps,
            **kw,
        )
        self.post_attention_layernorm = RMSNorm(p.hidden_size, eps=p.rms_norm_eps, **kw)
        self.mlp = MoELayer(
            hidden_size=p.hidden_size,
            intermediate_size=p.intermediate_size,
            num_experts=p.num_experts,
            top_k=p.experts_top_k,
            use_expert_bias=p.use_expert_bias,
            shared_expert_intermediate_size=p.shared_expert_intermediate_size,
            **kw,
        )

    def reset_parameters(self) -> None:
        self.input_layernorm.reset_parameters()
        self.self_attn.reset_parameters()
        self.post_attention_layernorm.reset_parameters()
        self.mlp.reset_parameters()

    def forward(self, hidden_states: torch.Tensor, rotary_cos_sin) -> torch.Tensor:
        hidden_states = hidden_states + self.self_attn(
            self.input_layernorm(hidden_states), rotary_cos_sin
        )
        hidden_states = hidden_states + self.mlp(
            self.post_attention_layernorm(hidden_states)
        )
        return hidden_states


class Qwen3MoEModel(nn.Module):
    """Backbone (embed -> layers[start:end] -> norm), PP-sliceable."""

    def __init__(
        self,
        params: Qwen3MoEModelParameters,
        stage_info: PipelineStageInfo | None = None,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        self.params = params
        self.stage_info = stage_info or PipelineStageInfo(0, 1)
        kw = {"device": device, "dtype": dtype}

        start, end = self.stage_info.layer_range(
            params.num_hidden_layers,
            params.pipeline_num_virtual_layers_pre,
            params.pipeline_num_virtual_layers_post,
        )
        self.layer_start, self.layer_end = start, end

        if self.stage_info.is_first_stage:
            self.embed_tokens = SplitTokenEmbeddings(
                dict(params.split_vocab_size),
                list(params.split_vocab_order),
                params.hidden_size,
                **kw,
            )
        else:
            self.embed_tokens = None

        self.layers = nn.ModuleDict(
            {
                str(i): Qwen3MoEDecoderLayer(params, **kw)
                for i in range(start, end)
            }
        )

        if self.stage_info.is_last_stage:
            self.norm = RMSNorm(params.hidden_size, eps=params.rms_norm_eps, **kw)
        else:
            self.norm = None

        self.rotary = RotaryEmbeddingProvider(
            rope_dim=params.head_dim, base=params.rope_base, device=device
        )

    def reset_parameters(self) -> None:
        if self.embed_tokens is not None:
            self.embed_tokens.reset_parameters()
        for layer in self.layers.values():
            layer.reset_parameters()
        if self.norm is not None:
            self.norm.reset_parameters()
        self.rotary.reset_parameters()

    def forward(
        self,
        input_ids: torch.Tensor | None = None,  # (B, S) on first stage
        hidden_states: torch.Tensor | None = None,  # (B, S, H) on later stages
        position_ids: torch.Tensor | None = None,  # (B, S)
    ) -> dict[str, torch.Tensor]:
        if self.embed_tokens is not None:
            assert input_ids is not None
            hidden_states = self.embed_tokens(input_ids)
        assert hidden_states is not None

        B, S, _ = hidden_states.shape
        if position_ids is None:
            # Under sequence parallelism the inter-block hidden states carry
            # only S/tp rows, but RoPE is applied inside the attention region
            # AFTER the all-gather: positions must cover the FULL sequence.
            S_full = (
                input_ids.shape[1]
                if input_ids is not None
                else S * getattr(self, "_d9d_sp_factor", 1)
            )
            # context parallelism: this rank holds the cp_rank-th contiguous
            # sequence chunk — positions offset to the global range
            cp_rank, _cp = getattr(self, "_d9d_cp", (0, 1))
            position_ids = (
                (torch.arange(S_full, device=hidden_states.device)
                 + cp_rank * S_full)
                .unsqueeze(0).expand(B, S_full)
            )
        rotary_cos_sin = self.rotary(position_ids)

        for i in range(self.layer_start, self.layer_end):
            layer = self.layers[str(i)]
            if self.params.checkpoint_layers and self.training:
                hidden_states = checkpoint(
                    layer, hidden_states, rotary_cos_sin, use_reentrant=False
                )
            else:
                hidden_states = layer(hidden_states, rotary_cos_sin)

        if self.norm is not None:
            hidden_states = self.norm(hidden_states)
        return {"hidden_states": hidden_states}

    def pipeline_input_names(self) -> set[str]:
        names = {"position_ids"}
        if self.stage_info.is_first_stage:
            names.add("input_ids")
        return names

    # -- pipelining shape inference (ModuleSupportsPipelining) ----------------

    def _stage_dtype(self) -> torch.dtype:
        # P2P buffers must match the dtype ACTUALLY sent: the model's
        # parameter dtype (a hardcoded bf16 here sized CPU/fp32 pipeline
        # recv buffers at half the sender's bytes -- found via the ws=2
        # PP trainer test).
        for p in self.parameters():
            return p.dtype
        return torch.bfloat16

    def infer_stage_inputs_from_pipeline_inputs(
        self, pipeline_inputs: dict[str, Any], num_microbatches: int
    ) -> dict[str, torch.Tensor]:
        if self.stage_info.is_first_stage:
            return {}
        ids = pipeline_inputs["input_ids"]
        B, S = ids.shape[0] // num_microbatches, ids.shape[1]
        # sequence parallelism: inter-stage hidden states carry S/tp rows
        S //= getattr(self, "_d9d_sp_factor", 1)
        return {
            "hidden_states": torch.empty(
                B, S, self.params.hidden_size, dtype=self._stage_dtype(),
                device="meta",
            )
        }

    def infer_stage_outputs_from_pipeline_inputs(
        self, pipeline_inputs: dict[str, Any], num_microbatches: int
    ) -> dict[str, torch.Tensor]:
        if self.stage_info.is_last_stage:
            return {}
        ids = pipeline_inputs["input_ids"]
        B, S = ids.shape[0] // num_microbatches, ids.shape[1]
        # sequence parallelism: inter-stage hidden states carry S/tp rows
        S //= getattr(self, "_d9d_sp_factor", 1)
        return {
            "hidden_states": torch.empty(
                B, S, self.params.hidden_size, dtype=self._stage_dtype(),
                device="meta",
            )
        }


class Qwen3MoEForCausalLM(nn.Module):
    """Backbone + split LM head; returns per-token logps via the fused CE op."""

    def __init__(
        self,
        params: Qwen3MoEModelParameters,
        stage_info: PipelineStageInfo | None = None,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        self.params = params
        self.model = Qwen3MoEModel(params, stage_info, device=device, dtype=dtype)
        if self.model.stage_info.is_last_stage:
            self.lm_head = SplitLanguageModellingHead(
                dict(params.split_vocab_size),
                list(params.split_vocab_order),
                params.hidden_size,
                device=device,
                dtype=dtype,
            )
        else:
            self.lm_head = None

    def reset_parameters(self) -> None:
        self.model.reset_parameters()
        if self.lm_head is not None:
            self.lm_head.reset_parameters()

    def init_weights(self) -> None:
        self.reset_parameters()

    @property
    def stage_info(self) -> PipelineStageInfo:
        return self.model.stage_info

    def pipeline_input_names(self) -> set[str]:
        names = self.model.pipeline_input_names()
        if self.model.stage_info.is_last_stage:
            names.add("labels")
        return names

    def forward(
        self,
        input_ids: torch.Tensor | None = None,
        hidden_states: torch.Tensor | None = None,
        position_ids: torch.Tensor | None = None,
        labels: torch.Tensor | None = None,
    ) -> dict[str, torch.Tensor]:
        out = self.model(
            input_ids=input_ids,
            hidden_states=hidden_states,
            position_ids=position_ids,
        )
        if self.lm_head is None:
            return out
        h = out["hidden_states"]
        assert labels is not None, "last stage needs labels for the fused CE head"
        logps = self.lm_head(h, labels)  # (B, S) log p(label), 0 at ignored
        n_valid = (labels != LM_IGNORE_INDEX).sum().clamp_min(1)
        loss = -(logps.sum() / n_valid)
        return {"logps": logps, "loss": loss.unsqueeze(0)}

    # -- pipelining shape inference -------------------------------------------

    def infer_stage_inputs_from_pipeline_inputs(self, pipeline_inputs, num_microbatches):
        return self.model.infer_stage_inputs_from_pipeline_inputs(
            pipeline_inputs, num_microbatches
        )

    def infer_stage_outputs_from_pipeline_inputs(self, pipeline_inputs, num_microbatches):
        return self.model.infer_stage_outputs_from_pipeline_inputs(
            pipeline_inputs, num_microbatches
        )
