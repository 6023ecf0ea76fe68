from .gated_deltanet import (
    CausalShortDepthwiseConv1d,
    GatedDeltaNet,
    LogSigmoidDecayGate,
    chunk_gated_delta_rule,
    step_gated_delta_rule,
)

__all__ = [
    "CausalShortDepthwiseConv1d",
    "GatedDeltaNet",
    "LogSigmoidDecayGate",
    "chunk_gated_delta_rule",
    "step_gated_delta_rule",
]
