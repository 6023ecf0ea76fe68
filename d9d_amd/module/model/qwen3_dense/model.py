"""Qwen3 dense model family (reference: d9d/module/model/qwen3_dense/model.py)."""

from typing import Any

import torch
from torch import nn
from torch.utils.checkpoint import checkpoint

from ....ops import LM_IGNORE_INDEX
from ....pipelining import PipelineStageInfo
from ...block.attention import GroupedQueryAttention
from ...block.embedding import SplitTokenEmbeddings
from ...block.ffn import SwiGLU
from ...block.head import ClassificationHead, EmbeddingHead, SplitLanguageModellingHead
from ...block.normalization import RMSNorm
from ...block.positional import RotaryEmbeddingProvider
from .params import Qwen3DenseModelParameters


class Qwen3DenseDecoderLayer(nn.Module):
    def __init__(self, p: Qwen3DenseModelParameters, device=None, dtype=None) -> None:
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.input_layernorm = RMSNorm(p.hidden_size, eps=p.rms_norm_eps, **kw)
        self.self_attn = GroupedQueryAttention(
            hidden_size=p.hidden_size,
            num_attention_heads=p.num_attention_heads,
            num_key_value_heads=p.num_key_value_heads,
            head_dim=p.head_dim,
            rms_norm_eps=p.rms_norm_eps,
            use_qk_norm=p.use_qk_norm,
            **kw,
        )
        self.post_attention_layernorm = RMSNorm(p.hidden_size, eps=p.rms_norm_eps, **kw)
        self.mlp = SwiGLU(p.hidden_size, p.intermediate_size, **kw)

    def reset_parameters(self) -> None:
        for child in self.children():
            child.reset_parameters()

    def forward(self, hidden_states: torch.Tensor, rotary_cos_sin) -> torch.Tensor:
        hidden_states = hidden_states + self.self_attn(
            self.input_layernorm(hidden_states), rotary_cos_sin
        )
        hidden_states = hidden_states + self.mlp(
            self.post_attention_layernorm(hidden_states)
        )
        return hidden_states


class Qwen3DenseModel(nn.Module):
    def __init__(
        self,
        params: Qwen3DenseModelParameters,
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

        self.embed_tokens = (
            SplitTokenEmbeddings(
                dict(params.split_vocab_size),
                list(params.split_vocab_order),
                params.hidden_size,
                **kw,
            )
            if self.stage_info.is_first_stage
            else None
        )
        self.layers = nn.ModuleDict(
            {str(i): Qwen3DenseDecoderLayer(params, **kw) for i in range(start, end)}
        )
        self.norm = (
            RMSNorm(params.hidden_size, eps=params.rms_norm_eps, **kw)
            if self.stage_info.is_last_stage
            else None
        )
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
        input_ids: torch.Tensor | None = None,
        hidden_states: torch.Tensor | None = None,
        position_ids: torch.Tensor | None = None,
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


class _DenseWithHead(nn.Module):
    """Shared plumbing for the head variants."""

    def __init__(self, params, stage_info, device=None, dtype=None) -> None:
        super().__init__()
        self.params = params
        self.model = Qwen3DenseModel(params, stage_info, device=device, dtype=dtype)

    @property
    def stage_info(self) -> PipelineStageInfo:
        return self.model.stage_info

    _last_stage_extra: frozenset[str] = frozenset()

    def pipeline_input_names(self) -> set[str]:
        names = self.model.pipeline_input_names()
        if self.model.stage_info.is_last_stage:
            names |= self._last_stage_extra
        return names

    def init_weights(self) -> None:
        self.reset_parameters()

    def infer_stage_inputs_from_pipeline_inputs(self, pipeline_inputs, num_microbatches):
        return self.model.infer_stage_inputs_from_pipeline_inputs(pipeline_inputs, num_microbatches)

    def infer_stage_outputs_from_pipeline_inputs(self, pipeline_inputs, num_microbatches):
        return self.model.infer_stage_outputs_from_pipeline_inputs(pipeline_inputs, num_microbatches)


class Qwen3DenseForCausalLM(_DenseWithHead):
    _last_stage_extra = frozenset({"labels"})

    def __init__(self, params, stage_info=None, device=None, dtype=None) -> None:
        super().__init__(params, stage_info, device=device, dtype=dtype)
        self.lm_head = (
            SplitLanguageModellingHead(
                dict(params.split_vocab_size),
                list(params.split_vocab_order),
                params.hidden_size,
                device=device,
                dtype=dtype,
            )
            if self.model.stage_info.is_last_stage
            else None
        )

    def reset_parameters(self) -> None:
        self.model.reset_parameters()
        if self.lm_head is not None:
            self.lm_head.reset_parameters()

    def forward(
        self,
        input_ids=None,
        hidden_states=None,
        position_ids=None,
        labels=None,
    ) -> dict[str, torch.Tensor]:
        out = self.model(
            input_ids=input_ids, hidden_states=hidden_states, position_ids=position_ids
        )
        if self.lm_head is None:
            return out
        assert labels is not None
        logps = self.lm_head(out["hidden_states"], labels)
        n_valid = (labels != LM_IGNORE_INDEX).sum().clamp_min(1)
        loss = -(logps.sum() / n_valid)
        return {"logps": logps, "loss": loss.unsqueeze(0)}


class Qwen3DenseForClassification(_DenseWithHead):
    _last_stage_extra = frozenset({"pooling_mask"})

    def __init__(self, params, stage_info=None, device=None, dtype=None) -> None:
        super().__init__(params, stage_info, device=device, dtype=dtype)
        self.head = (
            ClassificationHead(
                params.hidden_size, params.num_classes, device=device, dtype=dtype
            )
            if self.model.stage_info.is_last_stage
            else None
        )

    def reset_parameters(self) -> None:
        self.model.reset_parameters()
        if self.head is not None:
            self.head.reset_parameters()

    def forward(
        self, input_ids=None, hidden_states=None, position_ids=None, pooling_mask=None
    ) -> dict[str, torch.Tensor]:
        out = self.model(
            input_ids=input_ids, hidden_states=hidden_states, position_ids=position_ids
        )
        if self.head is None:
            return out
        if pooling_mask is None:
            pooling_mask = torch.ones(
                out["hidden_states"].shape[:2], device=out["hidden_states"].device
            )
        return {"logits": self.head(out["hidden_states"], pooling_mask)}


class Qwen3DenseForEmbedding(_DenseWithHead):
    _last_stage_extra = frozenset({"pooling_mask"})

    def __init__(self, params, stage_info=None, device=None, dtype=None) -> None:
        super().__init__(params, stage_info, device=device, dtype=dtype)
        self.head = (
            EmbeddingHead(
                params.hidden_size, params.embedding_size, device=device, dtype=dtype
            )
            if self.model.stage_info.is_last_stage
            else None
        )

    def reset_parameters(self) -> None:
        self.model.reset_parameters()
        if self.head is not None:
            self.head.reset_parameters()

    def forward(
        self, input_ids=None, hidden_states=None, position_ids=None, pooling_mask=None
    ) -> dict[str, torch.Tensor]:
        out = self.model(
            input_ids=input_ids, hidden_states=hidden_states, position_ids=position_ids
        )
        if self.head is None:
            return out
        if pooling_mask is None:
            pooling_mask = torch.ones(
                out["hidden_states"].shape[:2], device=out["hidden_states"].device
            )
        return {"embeddings": self.head(out["hidden_states"], pooling_mask)}
