"""Schedule factory (reference: d9d/pipelining/factory/).

Pydantic-discriminated schedule configs -> program builder -> executor.
`build_schedule` instantiates per-stage modules through `model_provider`
(callable taking PipelineStageInfo) and wraps them into PipelineStages.
"""

from dataclasses import dataclass
from typing import Annotated, Callable, Literal, Union

import torch
from pydantic import BaseModel, Field

from .actions import insert_communication_actions
from .api import PipelineStageInfo
from .executor import OfflinePipelineExecutor, PipelineScheduleExecutor
from .programs import (
    build_1f1b,
    build_looped_bfs,
    build_zb1p,
    build_dualpipev,
    build_zbv,
    local_stages,
    loop_stage_to_rank,
    v_stage_to_rank,
)
from .stage import PipelineStage


class PipelineScheduleInferenceConfig(BaseModel):
    schedule: Literal["inference"] = "inference"
    num_stages_per_rank: int = 1


class PipelineScheduleGPipeConfig(BaseModel):
    schedule: Literal["gpipe"] = "gpipe"


class PipelineScheduleLoopedBFSConfig(BaseModel):
    schedule: Literal["looped_bfs"] = "looped_bfs"
    num_stages_per_rank: int = 1


class PipelineSchedule1F1BConfig(BaseModel):
    schedule: Literal["1f1b"] = "1f1b"
    zero_bubble: bool = False
    num_stages_per_rank: int = 1  # > 1: interleaved (Megatron virtual pipeline)


class PipelineScheduleZB1PConfig(BaseModel):
    schedule: Literal["zb1p"] = "zb1p"


class PipelineScheduleZBVConfig(BaseModel):
    schedule: Literal["zero_bubble_v"] = "zero_bubble_v"


class PipelineScheduleDualPipeVConfig(BaseModel):
    """DualPipeV (DeepSeek bidirectional V schedule): interleaved F/B steady
    state with a zero-bubble weight-grad ramp in the drain. Requires 2 stages
    per rank and num_microbatches >= 2*pp. The reference runtime executes its
    F/B compose pairs sequentially too; communication overlap comes from the
    batched async P2P layer."""

    schedule: Literal["dual_pipe_v"] = "dual_pipe_v"


PipelineScheduleConfig = Annotated[
    Union[
        PipelineScheduleInferenceConfig,
        PipelineScheduleGPipeConfig,
        PipelineScheduleLoopedBFSConfig,
        PipelineSchedule1F1BConfig,
        PipelineScheduleZB1PConfig,
        PipelineScheduleZBVConfig,
        PipelineScheduleDualPipeVConfig,
    ],
    Field(discriminator="schedule"),
]


@dataclass
class PipelineScheduleInfo:
    schedule: object  # PipelineScheduleExecutor | OfflinePipelineExecutor
    stages: list[PipelineStage]
    modules: list[torch.nn.Module]
    has_first_stage: bool
    has_last_stage: bool


_V_SCHEDULES = ("zero_bubble_v", "dual_pipe_v")


def _num_stages(config, pp: int) -> int:
    if config.schedule in _V_SCHEDULES:
        return 2 * pp
    per_rank = getattr(config, "num_stages_per_rank", 1)
    return pp * per_rank


def build_schedule(
    config,
    model_provider: Callable[[PipelineStageInfo], torch.nn.Module],
    num_microbatches: int,
    device: torch.device,
    pp_rank: int = 0,
    pp_size: int = 1,
    pp_group=None,
    input_spec=None,
) -> PipelineScheduleInfo:
    num_stages = _num_stages(config, pp_size)
    if config.schedule in _V_SCHEDULES and pp_size > 1:
        rank_of_stage = v_stage_to_rank(num_stages, pp_size)
    else:
        rank_of_stage = loop_stage_to_rank(num_stages, pp_size)
    owned = local_stages(rank_of_stage, pp_rank)

    modules = []
    stages = []
    for g in owned:
        info = PipelineStageInfo(stage_index=g, num_stages=num_stages)
        module = model_provider(info)
        modules.append(module)
        stages.append(PipelineStage(module, g, num_stages, device))

    forward_only = config.schedule == "inference"
    if pp_size == 1:
        schedule = OfflinePipelineExecutor(
            stages, num_microbatches, input_spec, forward_only=forward_only
        )
        return PipelineScheduleInfo(schedule, stages, modules, True, True)

    name = config.schedule
    if name in ("gpipe", "looped_bfs", "inference"):
        prog = build_looped_bfs(pp_rank, pp_size, num_stages, num_microbatches, forward_only)
    elif name == "1f1b":
        prog = build_1f1b(pp_rank, pp_size, num_stages, num_microbatches,
                          zero_bubble=config.zero_bubble)
    elif name == "zb1p":
        prog = build_zb1p(pp_rank, pp_size, num_stages, num_microbatches)
    elif name == "zero_bubble_v":
        prog = build_zbv(pp_rank, pp_size, num_stages, num_microbatches)
    elif name == "dual_pipe_v":
        prog = build_dualpipev(pp_rank, pp_size, num_stages, num_microbatches)
    else:
        raise ValueError(f"unknown schedule {name!r}")

    stage_is_first = [g == 0 for g in owned]
    stage_is_last = [g == num_stages - 1 for g in owned]
    prev_is_local = [g > 0 and rank_of_stage[g - 1] == pp_rank for g in owned]
    next_is_local = [
        g < num_stages - 1 and rank_of_stage[g + 1] == pp_rank for g in owned
    ]
    prog = insert_communication_actions(
        prog, stage_is_first, stage_is_last, prev_is_local, next_is_local
    )

    schedule = PipelineScheduleExecutor(
        stages=stages,
        global_stage_ids=owned,
        rank_of_stage=rank_of_stage,
        program=prog,
        num_microbatches=num_microbatches,
        pp_rank=pp_rank,
        group=pp_group,
        input_spec=input_spec,
        forward_only=forward_only,
    )
    return PipelineScheduleInfo(
        schedule, stages, modules,
        schedule.has_first_stage, schedule.has_last_stage,
    )
