"""SDPA backend family (reference: d9d/module/block/attention/sdpa/, DEP-0008).

The MI355X framework has exactly ONE production backend — the CDNA4 flash
kernel — plus the eager fp32 oracle used as the numerics reference in tests.
The selection machinery (protocol, discriminated config, env override) is
kept so user configs stay source-compatible with the reference:
resolution order = explicit config > D9D_BACKEND_AUTO_SDPA env > auto.
"""

import os
from typing import Annotated, Literal, Protocol, Union, runtime_checkable

import torch
from pydantic import BaseModel, Field

from ....ops.attention import _eager_attention, flash_attn_func


@runtime_checkable
class SdpaBackend(Protocol):
    def __call__(
        self,
        q: torch.Tensor,
        k: torch.Tensor,
        v: torch.Tensor,
        *,
        causal: bool = True,
        softmax_scale: float | None = None,
        window_size: tuple[int, int] = (-1, -1),
        sinks: torch.Tensor | None = None,
        q_offset: int = 0,
    ) -> torch.Tensor: ...


class Cdna4FlashSdpaConfig(BaseModel):
    backend: Literal["cdna4_flash"] = "cdna4_flash"


class EagerSdpaConfig(BaseModel):
    backend: Literal["eager"] = "eager"


class TorchSdpaConfig(BaseModel):
    """torch.nn.functional.scaled_dot_product_attention (reference
    TorchSdpa, sdpa/impl/torch_sdpa.py): no sink or q_offset support, but
    exercises torch's own fused path (AOTriton/MIOpen on ROCm)."""

    backend: Literal["torch"] = "torch"


class AutoSdpaConfig(BaseModel):
    backend: Literal["auto"] = "auto"


SdpaBackendConfig = Annotated[
    Union[Cdna4FlashSdpaConfig, EagerSdpaConfig, TorchSdpaConfig, AutoSdpaConfig],
    Field(discriminator="backend"),
]

SDPA_ENV_VAR = "D9D_BACKEND_AUTO_SDPA"


def _cdna4_flash(q, k, v, *, causal=True, softmax_scale=None,
                 window_size=(-1, -1), sinks=None, q_offset=0):
    return flash_attn_func(
        q, k, v, causal=causal, softmax_scale=softmax_scale,
        window_size=window_size, sinks=sinks, q_offset=q_offset,
    )


def _eager(q, k, v, *, causal=True, softmax_scale=None,
           window_size=(-1, -1), sinks=None, q_offset=0):
    import math

    scale = softmax_scale or 1.0 / math.sqrt(q.shape[-1])
    out, _ = _eager_attention(q, k, v, causal, scale, window_size, sinks, q_offset)
    return out


def _torch_sdpa(q, k, v, *, causal=True, softmax_scale=None,
                window_size=(-1, -1), sinks=None, q_offset=0):
    if sinks is not None or q_offset != 0:
        raise ValueError("torch sdpa backend does not support sinks/q_offset")
    import torch.nn.functional as F

    # (B, S, H, D) -> (B, H, S, D)
    qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
    kw = {}
    if window_size != (-1, -1):
        # explicit mask: sliding window + causal
        S, Skv = q.shape[1], k.shape[1]
        pos_q = torch.arange(S, device=q.device).unsqueeze(1)
        pos_k = torch.arange(Skv, device=q.device).unsqueeze(0)
        off = Skv - S
        mask = torch.ones(S, Skv, dtype=torch.bool, device=q.device)
        if causal:
            mask &= pos_k <= pos_q + off
        if window_size[0] >= 0:
            mask &= pos_k >= pos_q + off - window_size[0]
        out = F.scaled_dot_product_attention(
            qt, kt, vt, attn_mask=mask, scale=softmax_scale, enable_gqa=True
        )
    else:
        out = F.scaled_dot_product_attention(
            qt, kt, vt, is_causal=causal, scale=softmax_scale, enable_gqa=True,
            **kw,
        )
    return out.transpose(1, 2)


def build_sdpa_backend(config: SdpaBackendConfig | None = None) -> SdpaBackend:
    name = None
    if config is not None and config.backend != "auto":
        name = config.backend
    elif SDPA_ENV_VAR in os.environ:
        name = os.environ[SDPA_ENV_VAR]
    if name == "eager":
        return _eager
    if name == "torch":
        return _torch_sdpa
    # auto and cdna4_flash both resolve to the flash op (which itself keeps
    # the eager path for CPU and for features the kernel lacks).
    return _cdna4_flash
