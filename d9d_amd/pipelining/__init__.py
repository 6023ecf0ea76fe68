from .api import (
    PipelineStageInfo,
    distribute_layers_for_pipeline_stage,
    ModuleSupportsPipelining,
)

__all__ = [
    "PipelineStageInfo",
    "distribute_layers_for_pipeline_stage",
    "ModuleSupportsPipelining",
]
