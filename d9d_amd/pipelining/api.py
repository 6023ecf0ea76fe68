"""Pipeline-facing model API (reference: d9d/pipelining/api/module.py:8-102).

`PipelineStageInfo` identifies one virtual stage; models use
`distribute_layers_for_pipeline_stage` to slice their global layer index
range, with virtual pre/post layers accounting for embedding/head cost.
`ModuleSupportsPipelining` is the shape-inference protocol the stage buffer
allocator uses (run on the meta device).
"""

from dataclasses import dataclass
from typing import Any, Protocol, runtime_checkable

import torch


@dataclass(frozen=True)
class PipelineStageInfo:
    stage_index: int
    num_stages: int

    @property
    def is_first_stage(self) -> bool:
        return self.stage_index == 0

    @property
    def is_last_stage(self) -> bool:
        return self.stage_index == self.num_stages - 1

    def layer_range(
        self,
        num_layers: int,
        num_virtual_pre: int = 0,
        num_virtual_post: int = 0,
    ) -> tuple[int, int]:
        return distribute_layers_for_pipeline_stage(
            num_layers, self.stage_index, self.num_stages,
            num_virtual_pre, num_virtual_post,
        )


def distribute_layers_for_pipeline_stage(
    num_layers: int,
    stage_index: int,
    num_stages: int,
    num_virtual_layers_pre: int = 0,
    num_virtual_layers_post: int = 0,
) -> tuple[int, int]:
    """[start, end) of global layer indices owned by `stage_index`.

    Virtual layers model embedding (pre) and head (post) cost: they are
    added to the balance computation but never returned
    (reference: api/module.py:38-98).
    """
    total = num_layers + num_virtual_layers_pre + num_virtual_layers_post
    base, rem = divmod(total, num_stages)
    # Stages at the END get the extra layers (the first stages already carry
    # the virtual-pre weight; mirrors the reference's balancing).
    bounds = [0]
    for s in range(num_stages):
        size = base + (1 if s >= num_stages - rem else 0)
        bounds.append(bounds[-1] + size)
    start_v, end_v = bounds[stage_index], bounds[stage_index + 1]
    # Strip the virtual layers back out.
    start = max(start_v - num_virtual_layers_pre, 0)
    end = min(end_v - num_virtual_layers_pre, num_layers)
    end = max(end, start)
    return start, end


@runtime_checkable
class ModuleSupportsPipelining(Protocol):
    """Shape inference for P2P buffer allocation, run on the meta device."""

    def infer_stage_inputs_from_pipeline_inputs(
        self,
        pipeline_inputs: dict[str, Any],
        num_microbatches: int,
    ) -> dict[str, torch.Tensor]: ...

    def infer_stage_outputs_from_pipeline_inputs(
        self,
        pipeline_inputs: dict[str, Any],
        num_microbatches: int,
    ) -> dict[str, torch.Tensor]: ...
