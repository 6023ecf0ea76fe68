from .grouped_query import GroupedQueryAttention
from .multi_head_latent import MultiHeadLatentAttention, LowRankProjection
from .sdpa import SdpaBackend, SdpaBackendConfig, build_sdpa_backend
from .linear.gated_deltanet import GatedDeltaNet, chunk_gated_delta_rule

__all__ = [
    "GroupedQueryAttention",
    "MultiHeadLatentAttention",
    "LowRankProjection",
    "SdpaBackend",
    "SdpaBackendConfig",
    "build_sdpa_backend",
    "GatedDeltaNet",
    "chunk_gated_delta_rule",
]
